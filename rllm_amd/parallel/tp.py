"""Tensor-parallel sharding for the rollout engine (BASELINE config 5:
TP=2 rollout for the 14B model; reference knob
rollout.tensor_model_parallel_size, _generated yaml:85).

Sharding plan (Megatron-style, attention-head aligned):
  qkv_proj      column-parallel by head: rank r keeps q heads
                [r*Hq/tp, (r+1)*Hq/tp) and kv heads [r*Hk/tp, ...)
  o_proj        row-parallel (input = this rank's q heads) -> all_reduce
  gate_up_proj  column-parallel (I/tp per half, gate and up sharded alike)
  down_proj     row-parallel -> all_reduce
  embed/lm_head replicated (vocab-sharded logits would force an
                all-gather before sampling; replication keeps sampling
                deterministic and identical on every rank, so all TP
                ranks can run the scheduler in lockstep with no extra
                coordination)
  KV cache      per-rank Hk/tp heads (cache memory divides by tp)

Two all-reduces of [T, H] bf16 per layer over xGMI. The sharding math
here is pure tensor slicing — CPU-testable; the collectives ride
torch.distributed (RCCL on GPU, gloo in tests).
"""

from __future__ import annotations

import torch

from rllm_amd.models.config import ModelConfig


def shard_model_config(cfg: ModelConfig, tp: int) -> ModelConfig:
    assert cfg.num_heads % tp == 0, f"num_heads {cfg.num_heads} not divisible by tp={tp}"
    assert cfg.num_kv_heads % tp == 0, f"num_kv_heads {cfg.num_kv_heads} not divisible by tp={tp}"
    assert cfg.intermediate_size % tp == 0
    return ModelConfig(
        name=f"{cfg.name}-tp{tp}",
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size // tp,
        num_layers=cfg.num_layers,
        num_heads=cfg.num_heads // tp,
        num_kv_heads=cfg.num_kv_heads // tp,
        head_dim=cfg.head_dim,
        vocab_size=cfg.vocab_size,
        rope_theta=cfg.rope_theta,
        rms_eps=cfg.rms_eps,
        tie_word_embeddings=cfg.tie_word_embeddings,
        max_position_embeddings=cfg.max_position_embeddings,
        qkv_bias=cfg.qkv_bias,
    )


def shard_qkv(qkv_weight: torch.Tensor, cfg: ModelConfig, rank: int, tp: int) -> torch.Tensor:
    """[ (Hq+2Hk)*D, H ] -> this rank's head slices, keeping q|k|v layout."""
    D = cfg.head_dim
    q, k, v = torch.split(qkv_weight, [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=0)
    hq, hk = cfg.num_heads // tp, cfg.num_kv_heads // tp
    qs = q[rank * hq * D : (rank + 1) * hq * D]
    ks = k[rank * hk * D : (rank + 1) * hk * D]
    vs = v[rank * hk * D : (rank + 1) * hk * D]
    return torch.cat([qs, ks, vs], dim=0)


def shard_qkv_bias(bias: torch.Tensor, cfg: ModelConfig, rank: int, tp: int) -> torch.Tensor:
    return shard_qkv(bias.unsqueeze(-1), cfg, rank, tp).squeeze(-1)


def shard_o(o_weight: torch.Tensor, cfg: ModelConfig, rank: int, tp: int) -> torch.Tensor:
    """[H, q_size] row-parallel on the input dim (this rank's q heads)."""
    D = cfg.head_dim
    hq = cfg.num_heads // tp
    return o_weight[:, rank * hq * D : (rank + 1) * hq * D].contiguous()


def shard_gate_up(w: torch.Tensor, cfg: ModelConfig, rank: int, tp: int) -> torch.Tensor:
    """[2I, H] with gate|up halves each column-sharded."""
    I = cfg.intermediate_size
    gate, up = w[:I], w[I:]
    s = I // tp
    return torch.cat([gate[rank * s : (rank + 1) * s], up[rank * s : (rank + 1) * s]], dim=0)


def shard_down(w: torch.Tensor, cfg: ModelConfig, rank: int, tp: int) -> torch.Tensor:
    """[H, I] row-parallel on the input dim."""
    I = cfg.intermediate_size
    s = I // tp
    return w[:, rank * s : (rank + 1) * s].contiguous()


def shard_state_dict(full_sd: dict[str, torch.Tensor], cfg: ModelConfig,
                     rank: int, tp: int) -> dict[str, torch.Tensor]:
    """Shard a QwenModel state dict (fused layout) for one TP rank."""
    out: dict[str, torch.Tensor] = {}
    for name, w in full_sd.items():
        if name.endswith("qkv_proj"):
            out[name] = shard_qkv(w, cfg, rank, tp)
        elif name.endswith("qkv_bias"):
            out[name] = shard_qkv_bias(w, cfg, rank, tp)
        elif name.endswith("o_proj"):
            out[name] = shard_o(w, cfg, rank, tp)
        elif name.endswith("gate_up_proj"):
            out[name] = shard_gate_up(w, cfg, rank, tp)
        elif name.endswith("down_proj"):
            out[name] = shard_down(w, cfg, rank, tp)
        else:  # norms, embeddings, lm_head: replicated
            out[name] = w
    return out


def tp_reference_forward(full_weight_matmul, shards: list[torch.Tensor], x: torch.Tensor,
                         mode: str) -> torch.Tensor:
    """Testing helper: simulate the TP collective on CPU.
    column: concat of per-rank outputs == full output.
    row:    sum of per-rank partial outputs == full output (the all_reduce)."""
    if mode == "column":
        return torch.cat([x @ s.t() for s in shards], dim=-1)
    if mode == "row":
        outs = None
        off = 0
        for s in shards:
            xi = x[..., off : off + s.shape[1]]
            off += s.shape[1]
            o = xi @ s.t()
            outs = o if outs is None else outs + o
        return outs
    raise ValueError(mode)


# ---------------------------------------------------------------------------
# Runtime wiring (rollout TP): sharded model construction + per-layer
# all-reduce hooks. The forward hooks live in models/qwen.py (`tp_group` on
# each layer: o_proj and down_proj partials are summed over xGMI).
# ---------------------------------------------------------------------------


def build_tp_model(cfg: ModelConfig, tp_rank: int, tp: int, *, device: str = "cuda",
                   seed: int = 0, tp_group=None, full_state_dict: dict | None = None):
    """One TP rank's QwenModel: identical full-model init on every rank
    (same seed) sliced to this rank's shard, so all ranks agree on the
    replicated tensors bit-for-bit (embeddings/lm_head/norms — required
    for lockstep sampling)."""
    from rllm_amd.models.qwen import QwenModel

    if full_state_dict is None:
        full = QwenModel(cfg, device="cpu").init_random(seed=seed)
        full_state_dict = full.state_dict()
    shard_cfg = shard_model_config(cfg, tp)
    sd = shard_state_dict(full_state_dict, cfg, tp_rank, tp)
    model = QwenModel(shard_cfg, device=device)
    model.load_state_dict({k: v.to(device) for k, v in sd.items()})
    enable_tp(model, tp_group)
    return model


def enable_tp(model, tp_group) -> None:
    """Arm the per-layer all-reduce hooks (rollout paths only: the hooks
    fire in _finish / forward_decode_fused; the training path must not run
    under TP — gradients would need their own collective plan)."""
    model.tp_group = tp_group
    for layer in model.layers:
        layer.tp_group = tp_group
