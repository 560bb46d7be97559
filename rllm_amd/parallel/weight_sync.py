"""Trainer→rollout weight synchronization.

Two modes:

* COLOCATED (default): the rollout engine shares the actor's weight
  tensors on the same GPU — sync is a version stamp, zero bytes moved
  (what the reference needs a cupy-NCCL checkpoint engine for,
  verl_backend.py:362-388).

* SEPARATED: trainer rank broadcasts the flat bf16 parameter buffer over
  the collective group to standalone rollout ranks, following the
  reference's pause → drain → version++ → broadcast → resume protocol
  (unified_trainer.py:812-827, param_sync.py:93-172). The flat buffer
  means ONE large collective instead of per-tensor chatter — sized for
  xGMI's per-link bandwidth rather than many small messages.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


@dataclass
class WeightSyncGroup:
    """A collective group spanning the trainer rank + rollout ranks."""

    group: object  # torch.distributed ProcessGroup
    src_rank: int  # global rank of the trainer (broadcast source)


class ColocatedWeightSync:
    """Zero-copy: engine tensors ARE trainer tensors."""

    def __init__(self, engine=None, gateway=None):
        self.engine = engine
        self.gateway = gateway
        self.version = 0

    async def sync(self, version: int) -> None:
        self.version = version
        if self.engine is not None:
            self.engine.weight_version = version
        if self.gateway is not None:
            await self.gateway.aset_weight_version(version)


class SeparatedWeightSync:
    """Flat-buffer broadcast across a trainer+rollout process group."""

    def __init__(self, flat_param: torch.Tensor, sync_group: WeightSyncGroup,
                 engine=None):
        self.flat_param = flat_param
        self.sync_group = sync_group
        self.engine = engine
        self.version = 0

    def sync(self, version: int, pause_resume: bool = True) -> None:
        eng = self.engine
        if pause_resume and eng is not None:
            eng.pause()
        dist.broadcast(self.flat_param, src=self.sync_group.src_rank,
                       group=self.sync_group.group)
        self.version = version
        if eng is not None:
            eng.weight_version = version
            if pause_resume:
                eng.resume()


def broadcast_state_dict(module: torch.nn.Module, src: int = 0, group=None) -> None:
    """Per-tensor broadcast fallback (checkpoint-shaped models without a
    flat buffer)."""
    for p in module.state_dict().values():
        if isinstance(p, torch.Tensor):
            dist.broadcast(p, src=src, group=group)
