"""HF checkpoint importer tests (reference loads real models via
tokenizer/processor + verl weight init, train_agent_ppo.py:108-126).

A tiny Qwen2 checkpoint is built OFFLINE with transformers and imported
through rllm_amd.models.hf_loader; CPU tests verify config parsing, the
fused-layout weight mapping, TP shard reassembly, and logits parity of a
pure-torch mirror of our architecture against transformers fp32. The GPU
test (marked) checks the real HIP forward against transformers."""

import json

import pytest
import torch

from rllm_amd.models.config import ModelConfig
from rllm_amd.models.hf_loader import (
    LazyHFStateDict,
    config_from_hf,
    is_hf_model_dir,
    load_hf_model,
)

transformers = pytest.importorskip("transformers")


TINY = dict(hidden_size=64, intermediate_size=128, num_hidden_layers=2,
            num_attention_heads=4, num_key_value_heads=2, vocab_size=320,
            max_position_embeddings=512, tie_word_embeddings=False,
            rope_theta=10000.0, rms_norm_eps=1e-6)


@pytest.fixture(scope="module")
def hf_dir(tmp_path_factory):
    from transformers import Qwen2Config, Qwen2ForCausalLM

    torch.manual_seed(7)
    d = tmp_path_factory.mktemp("tiny_qwen2_hf")
    m = Qwen2ForCausalLM(Qwen2Config(**TINY))
    # quantize to bf16 and back so our bf16 import starts from identical values
    m = m.to(torch.bfloat16).to(torch.float32)
    m.save_pretrained(str(d), safe_serialization=True)
    return str(d)


def test_is_hf_model_dir(hf_dir, tmp_path):
    assert is_hf_model_dir(hf_dir)
    assert not is_hf_model_dir(tmp_path)
    assert not is_hf_model_dir("r1-distill-qwen-1.5b")


def test_config_from_hf(hf_dir):
    cfg = config_from_hf(hf_dir)
    assert isinstance(cfg, ModelConfig)
    assert cfg.hidden_size == 64
    assert cfg.intermediate_size == 128
    assert cfg.num_layers == 2
    assert cfg.num_heads == 4 and cfg.num_kv_heads == 2
    assert cfg.head_dim == 16
    assert cfg.vocab_size == 320
    assert cfg.tie_word_embeddings is False
    assert cfg.rope_theta == 10000.0


def test_config_rejects_unknown_arch(hf_dir, tmp_path):
    raw = json.loads(open(f"{hf_dir}/config.json").read())
    raw["architectures"] = ["LlamaForCausalLM"]
    (tmp_path / "config.json").write_text(json.dumps(raw))
    with pytest.raises(ValueError, match="unsupported architecture"):
        config_from_hf(tmp_path)


def test_weight_mapping_fused_layout(hf_dir):
    """qkv/gate_up fusions match manual concat of the HF tensors."""
    model, cfg = load_hf_model(hf_dir, device="cpu")
    sd = LazyHFStateDict(hf_dir)
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        want_qkv = torch.cat([sd[p + "self_attn.q_proj.weight"],
                              sd[p + "self_attn.k_proj.weight"],
                              sd[p + "self_attn.v_proj.weight"]], dim=0).to(torch.bfloat16)
        assert torch.equal(layer.qkv_proj.detach(), want_qkv)
        want_b = torch.cat([sd[p + "self_attn.q_proj.bias"],
                            sd[p + "self_attn.k_proj.bias"],
                            sd[p + "self_attn.v_proj.bias"]], dim=0).to(torch.bfloat16)
        assert torch.equal(layer.qkv_bias.detach(), want_b)
        want_gu = torch.cat([sd[p + "mlp.gate_proj.weight"],
                             sd[p + "mlp.up_proj.weight"]], dim=0).to(torch.bfloat16)
        assert torch.equal(layer.gate_up_proj.detach(), want_gu)
        assert torch.equal(layer.down_proj.detach(),
                           sd[p + "mlp.down_proj.weight"].to(torch.bfloat16))
        assert torch.equal(layer.o_proj.detach(),
                           sd[p + "self_attn.o_proj.weight"].to(torch.bfloat16))
    assert torch.equal(model.embed_tokens.detach(),
                       sd["model.embed_tokens.weight"].to(torch.bfloat16))
    assert torch.equal(model.lm_head.detach(), sd["lm_head.weight"].to(torch.bfloat16))


def test_tied_embeddings_handling(hf_dir, tmp_path):
    """A tied checkpoint (no lm_head.weight) loads with lm_head=None."""
    from transformers import Qwen2Config, Qwen2ForCausalLM

    torch.manual_seed(8)
    cfg = dict(TINY)
    cfg["tie_word_embeddings"] = True
    m = Qwen2ForCausalLM(Qwen2Config(**cfg))
    m.save_pretrained(str(tmp_path), safe_serialization=True)
    model, full = load_hf_model(tmp_path, device="cpu")
    assert model.lm_head is None
    assert torch.equal(model.lm_weight, model.embed_tokens)


def test_tp_shard_reassembly(hf_dir):
    """tp=2 shards of every layer tensor reassemble to the full fused one."""
    full, _ = load_hf_model(hf_dir, device="cpu")
    s0, c0 = load_hf_model(hf_dir, device="cpu", tp_rank=0, tp_size=2)
    s1, _ = load_hf_model(hf_dir, device="cpu", tp_rank=1, tp_size=2)
    cfg = full.cfg
    D = cfg.head_dim
    for lf, l0, l1 in zip(full.layers, s0.layers, s1.layers):
        # qkv: per-component head-slice concat
        q, k, v = torch.split(lf.qkv_proj, [cfg.q_size, cfg.kv_size, cfg.kv_size])
        q0, k0, v0 = torch.split(l0.qkv_proj, [cfg.q_size // 2, cfg.kv_size // 2, cfg.kv_size // 2])
        q1, k1, v1 = torch.split(l1.qkv_proj, [cfg.q_size // 2, cfg.kv_size // 2, cfg.kv_size // 2])
        assert torch.equal(torch.cat([q0, q1]), q)
        assert torch.equal(torch.cat([k0, k1]), k)
        assert torch.equal(torch.cat([v0, v1]), v)
        # o: column concat on input dim
        assert torch.equal(torch.cat([l0.o_proj, l1.o_proj], dim=1), lf.o_proj)
        # gate_up: per-half row concat
        I = cfg.intermediate_size
        g, u = lf.gate_up_proj[:I], lf.gate_up_proj[I:]
        g0, u0 = l0.gate_up_proj[: I // 2], l0.gate_up_proj[I // 2 :]
        g1, u1 = l1.gate_up_proj[: I // 2], l1.gate_up_proj[I // 2 :]
        assert torch.equal(torch.cat([g0, g1]), g)
        assert torch.equal(torch.cat([u0, u1]), u)
        assert torch.equal(torch.cat([l0.down_proj, l1.down_proj], dim=1), lf.down_proj)
    # replicated tensors
    assert torch.equal(s0.embed_tokens, full.embed_tokens)
    assert torch.equal(s1.lm_head, full.lm_head)


def _mirror_forward_fp32(model, ids: torch.Tensor) -> torch.Tensor:
    """Pure-torch fp32 forward over OUR fused weights mirroring the HIP
    path's math (rmsnorm → qkv+rope → causal GQA attention → swiglu),
    NeoX rotate-half RoPE exactly as ops/csrc/elementwise.hip:118-152."""
    cfg = model.cfg
    T = ids.numel()
    D, Hq, Hk = cfg.head_dim, cfg.num_heads, cfg.num_kv_heads
    h = model.embed_tokens.detach().float()[ids]
    pos = torch.arange(T)
    half = D // 2
    inv_freq = 1.0 / (cfg.rope_theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    freqs = torch.outer(pos.double(), inv_freq)
    cos, sin = freqs.cos().float(), freqs.sin().float()

    def rms(x, w):
        return x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + cfg.rms_eps) * w.float()

    def rope(x):  # [T, H, D]
        x1, x2 = x[..., :half], x[..., half:]
        c, s = cos[:, None, :], sin[:, None, :]
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)

    for layer in model.layers:
        x = rms(h, layer.input_layernorm.detach())
        qkv = x @ layer.qkv_proj.detach().float().t() + layer.qkv_bias.detach().float()
        q, k, v = torch.split(qkv, [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1)
        q = rope(q.view(T, Hq, D))
        k = rope(k.view(T, Hk, D))
        v = v.view(T, Hk, D)
        k = k.repeat_interleave(Hq // Hk, dim=1)
        v = v.repeat_interleave(Hq // Hk, dim=1)
        att = torch.einsum("thd,shd->hts", q, k) / D ** 0.5
        mask = torch.triu(torch.ones(T, T, dtype=torch.bool), diagonal=1)
        att = att.masked_fill(mask, float("-inf")).softmax(-1)
        o = torch.einsum("hts,shd->thd", att, v).reshape(T, Hq * D)
        h = h + o @ layer.o_proj.detach().float().t()
        x = rms(h, layer.post_attention_layernorm.detach())
        gu = x @ layer.gate_up_proj.detach().float().t()
        g, u = gu.chunk(2, dim=-1)
        h = h + (torch.nn.functional.silu(g) * u) @ layer.down_proj.detach().float().t()
    h = rms(h, model.norm.detach())
    return h @ model.lm_weight.detach().float().t()


def test_cpu_logits_parity_vs_transformers(hf_dir):
    """The imported weights drive a pure-torch mirror of our architecture
    to the SAME logits transformers fp32 produces — catches transposed
    maps, wrong RoPE convention, bias mis-fusing."""
    from transformers import Qwen2ForCausalLM

    hf = Qwen2ForCausalLM.from_pretrained(hf_dir, torch_dtype=torch.float32)
    hf.eval()
    model, _ = load_hf_model(hf_dir, device="cpu")

    torch.manual_seed(11)
    ids = torch.randint(0, 320, (24,))
    with torch.no_grad():
        want = hf(ids.unsqueeze(0)).logits[0]
        got = _mirror_forward_fp32(model, ids)
    lw = torch.log_softmax(want, -1)
    lg = torch.log_softmax(got, -1)
    # weights are bf16-quantized on our side; transformers holds fp32 of the
    # same (pre-quantized) values, so only op-order noise remains
    assert (lw - lg).abs().max().item() < 2e-2, (lw - lg).abs().max()
    # top-1 agreement everywhere
    assert (want.argmax(-1) == got.argmax(-1)).float().mean() > 0.95


@pytest.mark.gpu
def test_gpu_forward_train_parity_vs_transformers(tmp_path):
    """Real HIP forward_train on an imported checkpoint vs transformers
    fp32 logprobs (the VERDICT round-2 acceptance check)."""
    from transformers import Qwen2Config, Qwen2ForCausalLM

    torch.manual_seed(7)
    m = Qwen2ForCausalLM(Qwen2Config(**TINY))
    m = m.to(torch.bfloat16).to(torch.float32)
    m.save_pretrained(str(tmp_path), safe_serialization=True)
    m.eval()

    model, _ = load_hf_model(tmp_path, device="cuda")
    ids = torch.randint(0, 320, (32,), generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        want = torch.log_softmax(m(ids.unsqueeze(0)).logits[0].float(), -1)
        dev_ids = ids.to("cuda")
        posns = torch.arange(32, device="cuda", dtype=torch.int32)
        h = model.forward_train(dev_ids, posns, [0, 32])
        from rllm_amd import ops

        tgt = torch.roll(dev_ids, -1)
        lp, _ = ops.chunked_logprob(h, model.lm_weight, tgt, want_entropy=False)
    want_lp = want[torch.arange(32), torch.roll(ids, -1)]
    # last row's "target" wraps — ignore it
    diff = (lp[:-1].cpu() - want_lp[:-1]).abs()
    assert diff.max().item() < 5e-2, diff.max()
