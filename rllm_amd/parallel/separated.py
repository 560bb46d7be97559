"""Separated-mode topology: trainer ranks + standalone rollout ranks
(reference: the trainer/rollout split behind verl's separated placement,
unified_trainer.py:812-827 + param_sync.py:93-172 — redesigned as plain
torch.distributed process groups over RCCL/xGMI, no Ray).

Layout over WORLD_SIZE = T + R ranks (one process per GPU):

* ranks [0, T)  — trainers: DP group for grad all-reduce.
* ranks [T, T+R) — rollout workers: each owns a full engine replica.
* sync group = {trainer 0} ∪ rollout ranks: ONE flat bf16 broadcast per
  weight version (xGMI-friendly: one large transfer per link, not
  per-tensor chatter).

Control plane: trainer 0 drives rollout workers with
``broadcast_object_list`` commands on the sync group (sync / generate /
stop); results come back with ``gather_object``. Rollout workers run
``RolloutWorker.serve()`` — a blocking command loop.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Any, Callable

import torch
import torch.distributed as dist

from rllm_amd.parallel.weight_sync import SeparatedWeightSync, WeightSyncGroup

logger = logging.getLogger(__name__)


@dataclass
class SeparatedTopology:
    """Role/rank bookkeeping + the process groups. Construct on EVERY rank
    (new_group is collective) after init_process_group."""

    n_trainers: int
    n_rollout: int
    rank: int = field(default=-1)
    trainer_group: Any = None
    sync_group: WeightSyncGroup | None = None

    def __post_init__(self):
        world = dist.get_world_size()
        if self.n_trainers + self.n_rollout != world:
            raise ValueError(
                f"topology {self.n_trainers}+{self.n_rollout} != world {world}")
        if self.rank < 0:
            self.rank = dist.get_rank()
        trainer_ranks = list(range(self.n_trainers))
        rollout_ranks = list(range(self.n_trainers, world))
        # both new_group calls are collective: every rank executes both
        self.trainer_group = dist.new_group(trainer_ranks)
        sync_pg = dist.new_group([0] + rollout_ranks)
        self.sync_group = WeightSyncGroup(group=sync_pg, src_rank=0)
        self.trainer_ranks = trainer_ranks
        self.rollout_ranks = rollout_ranks

    @property
    def is_trainer(self) -> bool:
        return self.rank < self.n_trainers

    @property
    def is_rollout(self) -> bool:
        return not self.is_trainer

    @property
    def on_sync_group(self) -> bool:
        return self.rank == 0 or self.is_rollout


# ---------------------------------------------------------------------------
# Control plane
# ---------------------------------------------------------------------------


def _command(topo: SeparatedTopology, cmd: dict | None) -> dict:
    """Broadcast one control message from trainer 0 across the sync group.
    Rollout ranks pass cmd=None and receive."""
    box = [cmd]
    dist.broadcast_object_list(box, src=0, group=topo.sync_group.group)
    return box[0]


class RolloutWorker:
    """Command loop on a rollout rank. `generate_fn(tasks) -> results` is
    the local engine entry (NativeRolloutEngine.generate in production;
    anything callable in tests)."""

    def __init__(self, topo: SeparatedTopology, flat_param: torch.Tensor,
                 generate_fn: Callable[[list], list], engine=None):
        assert topo.is_rollout, "RolloutWorker must run on a rollout rank"
        self.topo = topo
        self.sync = SeparatedWeightSync(flat_param, topo.sync_group, engine=engine)
        self.generate_fn = generate_fn

    def serve(self) -> None:
        while True:
            cmd = _command(self.topo, None)
            kind = cmd["cmd"]
            if kind == "stop":
                logger.info("rollout rank %d stopping", self.topo.rank)
                return
            if kind == "sync":
                self.sync.sync(cmd["version"], pause_resume=cmd.get("pause", True))
            elif kind == "generate":
                mine = cmd["shards"][self.topo.rank - self.topo.n_trainers]
                results = self.generate_fn(mine)
                dist.gather_object(results, None, dst=0, group=self.topo.sync_group.group)
            else:
                raise ValueError(f"unknown rollout command {kind!r}")


class SeparatedRolloutClient:
    """Trainer-0-side handle driving the rollout fleet. Non-zero trainer
    ranks never touch the sync group — they receive results via a regular
    broadcast on WORLD from trainer 0 (`scatter_results`)."""

    def __init__(self, topo: SeparatedTopology, flat_param: torch.Tensor | None):
        self.topo = topo
        self.flat_param = flat_param
        self.version = 0

    def sync_weights(self, version: int, pause: bool = True) -> None:
        if self.topo.rank != 0:
            return
        _command(self.topo, {"cmd": "sync", "version": version, "pause": pause})
        dist.broadcast(self.flat_param, src=0, group=self.topo.sync_group.group)
        self.version = version

    def generate(self, tasks: list) -> list:
        """Round-robin shard tasks over rollout ranks; returns the flat
        result list in task order. Call on trainer 0 only."""
        if self.topo.rank != 0:
            return []
        R = self.topo.n_rollout
        shards = [tasks[i::R] for i in range(R)]
        _command(self.topo, {"cmd": "generate", "shards": shards})
        gathered: list = [None] * (R + 1)
        dist.gather_object([], gathered, dst=0, group=self.topo.sync_group.group)
        # gathered[0] is trainer 0's own empty list; interleave back
        out: list = [None] * len(tasks)
        for r, results in enumerate(gathered[1:]):
            for j, item in enumerate(results):
                out[r + j * R] = item
        return out

    def stop(self) -> None:
        if self.topo.rank == 0:
            _command(self.topo, {"cmd": "stop"})
