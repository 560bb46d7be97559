"""Distributed helpers: one process per GPU, torch.distributed over RCCL
(backend "nccl" IS RCCL on ROCm) for GPU runs, gloo for CPU tests.

Collectives used:
  C1 gradient all-reduce (flat bf16 buffer)
  C2 weight broadcast trainer→rollout (separated mode)
  C3 all_reduce(MAX) of micro-batch counts (dynamic batching desync guard,
     reference patch.py:23-66)
  C4 trajectory/episode all-gather (object collective)
"""

from __future__ import annotations

import os
from datetime import timedelta

import torch
import torch.distributed as dist


def init_from_env(backend: str | None = None, timeout_s: int = 1800) -> tuple[int, int, int]:
    """Initialize process group from torchrun env vars. Returns
    (rank, world_size, local_rank). No-op single-process when WORLD_SIZE<=1."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size <= 1:
        return 0, 1, 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size,
                                timeout=timedelta(seconds=timeout_s))
    if torch.cuda.is_available():
        # modulo: a world-2 gloo smoke on a 1-GPU box shares device 0
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    return rank, world_size, local_rank


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


# Default process group for ALL helpers below. None = WORLD (the usual DP
# case). Separated placement sets this to the trainer subgroup so
# PolicyTrainer's collectives never involve rollout ranks (which sit in
# their own command loop).
_DEFAULT_GROUP = None


def set_default_group(group) -> None:
    global _DEFAULT_GROUP
    _DEFAULT_GROUP = group


def get_rank() -> int:
    return dist.get_rank(group=_DEFAULT_GROUP) if is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size(group=_DEFAULT_GROUP) if is_initialized() else 1


def barrier():
    if is_initialized():
        dist.barrier(group=_DEFAULT_GROUP)


def all_reduce_sum_(tensor: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=_DEFAULT_GROUP)
    return tensor


def all_reduce_max_(tensor: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        dist.all_reduce(tensor, op=dist.ReduceOp.MAX, group=_DEFAULT_GROUP)
    return tensor


def all_reduce_scalar(value: float, op: str = "sum") -> float:
    """Reduce a python scalar across ranks (cpu tensor on gloo, gpu on rccl)."""
    if not is_initialized():
        return value
    device = "cuda" if dist.get_backend() == "nccl" else "cpu"
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM,
                    group=_DEFAULT_GROUP)
    return float(t.item())


def broadcast_(tensor: torch.Tensor, src: int = 0) -> torch.Tensor:
    if is_initialized():
        dist.broadcast(tensor, src=src, group=_DEFAULT_GROUP)
    return tensor


def all_gather_object_list(obj) -> list:
    """C4: gather python objects (episodes/trajectories) from all ranks."""
    if not is_initialized():
        return [obj]
    out = [None] * dist.get_world_size(group=_DEFAULT_GROUP)
    dist.all_gather_object(out, obj, group=_DEFAULT_GROUP)
    return out


def destroy():
    if is_initialized():
        dist.destroy_process_group()
