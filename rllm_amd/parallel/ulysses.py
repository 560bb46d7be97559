"""Ulysses sequence parallelism for the training forward (SURVEY.md §2.D —
the reference's ulysses_sequence_parallel_size knob, verl actor config).

Long-update-context scaling: each SP rank holds a token shard of the packed
varlen batch. Around attention, two all-to-alls swap the sharding axis —
tokens→heads going in (each rank sees EVERY token for H/P q-heads, so causal
attention is exact), heads→tokens coming out. Everything outside attention
(norms, GEMMs, logprob head) stays local to the token shard, so memory per
GPU scales as T/P while attention math is unchanged.

xGMI fit: an all-to-all moves T*H*D*2/P bytes per rank per direction in P-1
point-to-point messages — for P≤8 on one node every pair has a direct link,
which is exactly the topology xGMI gives (no switch, 7 links/GPU).

GQA head mapping: with Hk kv heads and SP degree P, either Hk % P == 0
(each rank gets Hk/P kv heads) or P % Hk == 0 (kv heads are replicated
P/Hk× before the scatter; rank r then holds kv head r*Hk//P). Both keep the
q-group→kv-head mapping of the dense layout, so the same GQA attention
kernel runs unchanged on the shard.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def sp_world(group=None) -> int:
    return dist.get_world_size(group=group) if dist.is_initialized() else 1


def _a2a(x: torch.Tensor, group) -> torch.Tensor:
    """all_to_all_single over the leading [P, ...] axis (equal splits).
    Both buffers MUST be dense: the collective reads/writes flat memory, so
    a strided empty_like of a permuted view would be silently garbled.

    gloo has no CUDA all-to-all — stage through host there (test/fallback
    path only; RCCL runs it natively over xGMI)."""
    x = x.contiguous()
    if x.is_cuda and dist.get_backend(group) == "gloo":
        xc = x.cpu()
        out = torch.empty_like(xc)
        dist.all_to_all_single(out, xc, group=group)
        return out.to(x.device)
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    return out


def _seq_to_head(x: torch.Tensor, group) -> torch.Tensor:
    """[T/P, H, D] token-sharded → [T, H/P, D] head-sharded."""
    P = sp_world(group)
    t_loc, H, D = x.shape
    assert H % P == 0, f"heads {H} not divisible by SP degree {P}"
    x = x.reshape(t_loc, P, H // P, D).permute(1, 0, 2, 3).contiguous()
    out = _a2a(x, group)  # [P, T/P, H/P, D]; chunk p = rank p's tokens
    return out.reshape(P * t_loc, H // P, D)


def _head_to_seq(x: torch.Tensor, group) -> torch.Tensor:
    """[T, H/P, D] head-sharded → [T/P, H, D] token-sharded (inverse)."""
    P = sp_world(group)
    T, h_loc, D = x.shape
    assert T % P == 0
    t_loc = T // P
    x = x.reshape(P, t_loc, h_loc, D)
    out = _a2a(x, group)  # chunk p = my tokens' head-group p
    return out.permute(1, 0, 2, 3).reshape(t_loc, P * h_loc, D)


class _SeqToHead(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _seq_to_head(x, group)

    @staticmethod
    def backward(ctx, g):
        return _head_to_seq(g, ctx.group), None


class _HeadToSeq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _head_to_seq(x, group)

    @staticmethod
    def backward(ctx, g):
        return _seq_to_head(g, ctx.group), None


def seq_to_head(x: torch.Tensor, group=None) -> torch.Tensor:
    return _SeqToHead.apply(x, group)


def head_to_seq(x: torch.Tensor, group=None) -> torch.Tensor:
    return _HeadToSeq.apply(x, group)


def replicate_kv(k: torch.Tensor, factor: int) -> torch.Tensor:
    """[T, Hk, D] → [T, Hk*factor, D]; head j*factor+i is a copy of head j
    (keeps the GQA q-group→kv-head mapping under the scatter)."""
    if factor == 1:
        return k
    T, Hk, D = k.shape
    return k.unsqueeze(2).expand(T, Hk, factor, D).reshape(T, Hk * factor, D)


def ulysses_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      attn_fn, group=None):
    """Run `attn_fn(q_full, k_full, v_full) -> [T, Hq/P, D]` with full-sequence
    visibility from token-sharded q/k/v, returning the token-sharded result.

    q: [T/P, Hq, D]; k, v: [T/P, Hk, D]. attn_fn receives head-sharded
    full-length tensors and must be head-independent (flash / SDPA are).
    """
    P = sp_world(group)
    if P == 1:
        return attn_fn(q, k, v)
    Hk = k.shape[1]
    if Hk % P != 0:
        if P % Hk != 0:
            raise ValueError(f"kv heads {Hk} incompatible with SP degree {P}")
        f = P // Hk
        k, v = replicate_kv(k, f), replicate_kv(v, f)
    q_h = seq_to_head(q, group)
    k_h = seq_to_head(k, group)
    v_h = seq_to_head(v, group)
    out = attn_fn(q_h, k_h, v_h)
    return head_to_seq(out, group)


# ---------------------------------------------------------------------------
# Packed-batch sharding helpers (host side)
# ---------------------------------------------------------------------------


def shard_slice(total_tokens: int, rank: int, world: int) -> slice:
    """Token rows owned by `rank`; total MUST be divisible by world (pad the
    packed batch with a dummy masked sequence first — trainer/batch.py
    pack_rows pad_to_multiple)."""
    if total_tokens % world:
        raise ValueError(f"packed length {total_tokens} not divisible by SP {world}")
    per = total_tokens // world
    return slice(rank * per, (rank + 1) * per)
