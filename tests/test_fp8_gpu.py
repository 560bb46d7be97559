"""GPU: fp8 (OCP e4m3) rollout path — quantization accuracy, tuned fp8
GEMM parity, and the engine's fp8 decode mode end-to-end."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")

from rllm_amd import ops  # noqa: E402
from rllm_amd.models.config import ModelConfig  # noqa: E402

CFG = ModelConfig(name="fp8-tiny", hidden_size=512, intermediate_size=1024,
                  num_layers=2, num_heads=8, num_kv_heads=2, head_dim=128,
                  vocab_size=1024, tie_word_embeddings=False)


@requires_gpu
def test_fp8_quant_roundtrip():
    t = torch.randn(64, 256, device="cuda") * 3.0
    t8, s = ops.fp8_quant(t)
    assert t8.dtype == torch.float8_e4m3fn and s.numel() == 1
    deq = t8.float() * s
    # e4m3 has ~2 mantissa decimal digits; relative to the tensor amax
    assert (deq - t).abs().max() <= t.abs().max() * 0.07


@requires_gpu
def test_fp8_gemm_parity():
    M, N, K = 256, 2048, 1536
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    w = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    res = ops.pretune_fp8_decode_shapes([(M, N, K)], iters=10, verbose=False)
    if not res:
        pytest.skip("no valid fp8 hipblaslt algo on this stack")
    w8, sw = ops.fp8_quant(w)
    y = ops.fp8_linear(x, w8, sw)
    assert y.dtype == torch.bfloat16
    ref = torch.nn.functional.linear(x.float(), w.float())
    err = (y.float() - ref).abs().max().item()
    assert err <= ref.abs().max().item() * 0.08 + 0.3, err
    # untuned shape falls back to the bf16 weight
    x2 = torch.randn(33, K, device="cuda").to(torch.bfloat16)
    y2 = ops.fp8_linear(x2, w8, sw, w_bf16=w)
    assert torch.equal(y2, torch.nn.functional.linear(x2, w))


@requires_gpu
def test_engine_fp8_decode_mode():
    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.qwen import QwenModel

    model = QwenModel(CFG, device="cuda").init_random(seed=21)
    eng = LLMEngine(model, kv_budget_bytes=64 << 20, eos_token_id=None,
                    seed=5, fp8_decode=True)
    assert model.fp8_decode and hasattr(model.layers[0], "_fp8")

    prompts = [list(range(5 + i, 37 + i)) for i in range(4)]
    outs = eng.generate(prompts, SamplingParams(temperature=1.0, max_tokens=16))
    assert all(len(o.token_ids) == 16 for o in outs)
    assert all(all(torch.isfinite(torch.tensor(o.logprobs)).tolist()) for o in outs)

    # weight bump re-quantizes the fp8 copies
    old8 = model.layers[0]._fp8["qkv_proj"][0].clone()
    with torch.no_grad():
        model.layers[0].qkv_proj.mul_(1.5)
    eng.weight_version = 3
    assert not torch.equal(model.layers[0]._fp8["qkv_proj"][0], old8)

    # rollout-vs-bf16 drift is the TIS-corrected quantity; sanity-bound it
    base = QwenModel(CFG, device="cuda").init_random(seed=21)
    beng = LLMEngine(base, kv_budget_bytes=64 << 20, eos_token_id=None,
                     seed=5, fp8_decode=False)
    a = beng.generate([prompts[0]], SamplingParams(temperature=0.0, max_tokens=8))[0]
    model2 = QwenModel(CFG, device="cuda").init_random(seed=21)
    feng = LLMEngine(model2, kv_budget_bytes=64 << 20, eos_token_id=None,
                     seed=5, fp8_decode=True)
    b = feng.generate([prompts[0]], SamplingParams(temperature=0.0, max_tokens=8))[0]
    # greedy first token should agree on a sane quantization
    assert a.token_ids[0] == b.token_ids[0]


@requires_gpu
def test_fp8_linear_delayed_scaling_converges():
    """Delayed scaling: call 1 uses the bootstrap scale; by call 2 the scale
    reflects the true amax and outputs match the dynamic-quant path."""
    M, N, K = 256, 2048, 1536
    ops.pretune_fp8_decode_shapes([(M, N, K)], iters=5, verbose=False)
    if not ops.require_ext().hbl_fp8_has(M, N, K):
        pytest.skip("no valid fp8 algo")
    w = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    w8, sw = ops.fp8_quant(w)
    scale = torch.ones(1, device="cuda")
    amax = torch.zeros(1, device="cuda")
    x = (torch.randn(M, K, device="cuda") * 7).to(torch.bfloat16)
    # warms the scale (explicit per-call update in standalone use)
    ops.fp8_linear_delayed(x, w8, sw, scale, amax, w_bf16=w, update_scale=True)
    expect_scale = x.float().abs().amax() / 448.0
    assert abs(scale.item() - expect_scale.item()) < 2e-3 * max(1.0, expect_scale.item())
    y2 = ops.fp8_linear_delayed(x, w8, sw, scale, amax, w_bf16=w, update_scale=True)
    ref = ops.fp8_linear(x, w8, sw, w_bf16=w)
    err = (y2.float() - ref.float()).abs().max().item()
    assert err <= ref.float().abs().max().item() * 0.05 + 0.5, err


@requires_gpu
def test_fused_fp8_producers_match_unfused():
    """add_rmsnorm_fp8_ / swiglu_fp8 == (bf16 producer -> fp8_quant_delayed)
    bitwise: same scale, same e4m3 values, same accumulated amax."""
    torch.manual_seed(9)
    T, H, I = 64, 512, 1024
    h1 = (torch.randn(T, H, device="cuda") * 0.5).to(torch.bfloat16)
    h2 = h1.clone()
    delta = (torch.randn(T, H, device="cuda") * 0.1).to(torch.bfloat16)
    w = torch.rand(H, device="cuda").to(torch.bfloat16) + 0.5
    scale = torch.full((1,), 0.37, device="cuda")
    amax_a = torch.zeros(1, device="cuda")
    amax_b = torch.zeros(1, device="cuda")

    # unfused: bf16 add_rmsnorm then standalone delayed quant
    y_bf16 = ops.add_rmsnorm_(h1, delta.clone(), w, 1e-6)
    y8_ref = torch.empty(T, H, device="cuda", dtype=torch.float8_e4m3fn)
    from rllm_amd.ops import require_ext

    require_ext().fp8_quant_delayed(y_bf16.contiguous(), y8_ref, scale, amax_a)

    # fused
    y8 = ops.add_rmsnorm_fp8_(h2, delta.clone(), w, 1e-6, scale, amax_b)
    assert torch.equal(h1, h2)  # residual update identical
    # fused quantizes the fp32 pre-bf16-rounding value; allow 1 e4m3 step
    d = (y8.float() - y8_ref.float()).abs()
    assert (d <= (y8_ref.float().abs() * 0.13 + scale.item() * 0.02)).all(), d.max()
    # amax agrees to bf16 rounding
    assert abs(amax_a.item() - amax_b.item()) <= abs(amax_a.item()) * 0.01 + 1e-3

    gu = (torch.randn(T, 2 * I, device="cuda") * 0.8).to(torch.bfloat16)
    out_bf16 = ops.swiglu(gu)
    amax_a.zero_(), amax_b.zero_()
    y8_ref2 = torch.empty(T, I, device="cuda", dtype=torch.float8_e4m3fn)
    require_ext().fp8_quant_delayed(out_bf16.contiguous(), y8_ref2, scale, amax_a)
    y8_2 = ops.swiglu_fp8(gu, scale, amax_b)
    d2 = (y8_2.float() - y8_ref2.float()).abs()
    assert (d2 <= (y8_ref2.float().abs() * 0.13 + scale.item() * 0.02)).all(), d2.max()
    assert abs(amax_a.item() - amax_b.item()) <= abs(amax_a.item()) * 0.01 + 1e-3


@requires_gpu
def test_engine_fp8_fused_decode_logprob_sanity():
    """fp8 fused-producer decode stays close to bf16 decode logprobs."""
    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.qwen import QwenModel

    model = QwenModel(CFG, device="cuda").init_random(seed=21)
    eng_bf16 = LLMEngine(model, kv_budget_bytes=64 << 20, eos_token_id=None, seed=5)
    prompts = [list(range(5 + i, 37 + i)) for i in range(4)]
    sp = SamplingParams(temperature=0.0, max_tokens=12)  # greedy for comparability
    outs_a = eng_bf16.generate(prompts, sp)

    model2 = QwenModel(CFG, device="cuda").init_random(seed=21)
    eng_fp8 = LLMEngine(model2, kv_budget_bytes=64 << 20, eos_token_id=None,
                        seed=5, fp8_decode=True)
    # fused path requires all three producer sites armed
    assert all(n in model2.layers[0]._fp8 for n in ("qkv_proj", "gate_up_proj", "down_proj"))
    outs_b = eng_fp8.generate(prompts, sp)
    agree = sum(int(a == b) for oa, ob in zip(outs_a, outs_b)
                for a, b in zip(oa.token_ids, ob.token_ids))
    total = sum(len(o.token_ids) for o in outs_a)
    assert agree / total > 0.7, f"fp8 fused decode diverged: {agree}/{total} greedy agreement"
