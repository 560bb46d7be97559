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
