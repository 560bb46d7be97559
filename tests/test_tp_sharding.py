"""TP sharding math (CPU): column/row shards recombine to the full
computation — the invariant the TP=2 rollout relies on."""

import torch

from rllm_amd.models.config import get_model_config
from rllm_amd.parallel.tp import (
    shard_down,
    shard_gate_up,
    shard_model_config,
    shard_o,
    shard_qkv,
    shard_qkv_bias,
    shard_state_dict,
    tp_reference_forward,
)


def test_shard_model_config_14b_tp2():
    cfg = get_model_config("qwen2.5-14b")
    s = shard_model_config(cfg, 2)
    assert s.num_heads == 20 and s.num_kv_heads == 4
    assert s.intermediate_size == cfg.intermediate_size // 2
    assert s.hidden_size == cfg.hidden_size


def test_qkv_column_shard_recombines():
    torch.manual_seed(0)
    cfg = get_model_config("tiny")  # Hq=4, Hk=2, D=128, H=256
    tp = 2
    W = torch.randn((cfg.q_size + 2 * cfg.kv_size), cfg.hidden_size)
    x = torch.randn(3, cfg.hidden_size)
    full = x @ W.t()
    fq, fk, fv = torch.split(full, [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1)

    shards = [shard_qkv(W, cfg, r, tp) for r in range(tp)]
    scfg = shard_model_config(cfg, tp)
    for r in range(tp):
        out = x @ shards[r].t()
        q, k, v = torch.split(out, [scfg.q_size, scfg.kv_size, scfg.kv_size], dim=-1)
        D = cfg.head_dim
        hq, hk = scfg.num_heads, scfg.num_kv_heads
        assert torch.allclose(q, fq[:, r * hq * D : (r + 1) * hq * D])
        assert torch.allclose(k, fk[:, r * hk * D : (r + 1) * hk * D])
        assert torch.allclose(v, fv[:, r * hk * D : (r + 1) * hk * D])


def test_o_row_shard_all_reduce():
    torch.manual_seed(1)
    cfg = get_model_config("tiny")
    tp = 2
    W = torch.randn(cfg.hidden_size, cfg.q_size)
    attn = torch.randn(3, cfg.q_size)  # per-head outputs, head-ordered
    full = attn @ W.t()
    # per-rank partial: this rank's q-head slice of attn through its o shard
    partial_sum = None
    D = cfg.head_dim
    hq = cfg.num_heads // tp
    for r in range(tp):
        shard = shard_o(W, cfg, r, tp)
        xi = attn[:, r * hq * D : (r + 1) * hq * D]
        o = xi @ shard.t()
        partial_sum = o if partial_sum is None else partial_sum + o
    assert torch.allclose(partial_sum, full, atol=1e-5)


def test_mlp_shards_recombine():
    torch.manual_seed(2)
    cfg = get_model_config("tiny")
    tp = 2
    Wgu = torch.randn(2 * cfg.intermediate_size, cfg.hidden_size)
    Wd = torch.randn(cfg.hidden_size, cfg.intermediate_size)
    x = torch.randn(3, cfg.hidden_size)

    def swiglu(gu):
        g, u = gu.chunk(2, -1)
        return torch.nn.functional.silu(g) * u

    full = swiglu(x @ Wgu.t()) @ Wd.t()

    total = None
    for r in range(tp):
        gu_s = shard_gate_up(Wgu, cfg, r, tp)
        d_s = shard_down(Wd, cfg, r, tp)
        inter = swiglu(x @ gu_s.t())
        o = inter @ d_s.t()
        total = o if total is None else total + o
    assert torch.allclose(total, full, rtol=1e-4, atol=1e-2)


def test_shard_state_dict_shapes():
    cfg = get_model_config("tiny")
    sd = {
        "layers.0.qkv_proj": torch.randn(cfg.q_size + 2 * cfg.kv_size, cfg.hidden_size),
        "layers.0.qkv_bias": torch.randn(cfg.q_size + 2 * cfg.kv_size),
        "layers.0.o_proj": torch.randn(cfg.hidden_size, cfg.q_size),
        "layers.0.gate_up_proj": torch.randn(2 * cfg.intermediate_size, cfg.hidden_size),
        "layers.0.down_proj": torch.randn(cfg.hidden_size, cfg.intermediate_size),
        "norm": torch.ones(cfg.hidden_size),
        "embed_tokens": torch.randn(cfg.vocab_size, cfg.hidden_size),
    }
    scfg = shard_model_config(cfg, 2)
    out = shard_state_dict(sd, cfg, 1, 2)
    assert out["layers.0.qkv_proj"].shape == (scfg.q_size + 2 * scfg.kv_size, cfg.hidden_size)
    assert out["layers.0.qkv_bias"].shape == (scfg.q_size + 2 * scfg.kv_size,)
    assert out["layers.0.o_proj"].shape == (cfg.hidden_size, scfg.q_size)
    assert out["layers.0.gate_up_proj"].shape == (2 * scfg.intermediate_size, cfg.hidden_size)
    assert out["layers.0.down_proj"].shape == (cfg.hidden_size, scfg.intermediate_size)
    assert out["embed_tokens"].shape == sd["embed_tokens"].shape  # replicated
