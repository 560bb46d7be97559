"""Lockstep driver for tensor-parallel rollout (completes parallel/tp.py:
every TP rank owns a sharded QwenModel + its own LLMEngine slice of the KV
cache; the engines must take IDENTICAL scheduling decisions so the
per-layer all-reduces line up).

Determinism argument: LLMEngine.step() is a pure function of (waiting
order, running set, KV free list, sampling seed/step counter). All ranks
start empty with the same pool size and seed; if every rank applies the
same add_request/abort stream in the same order, every subsequent
decision — prefill batch composition, preemption victims, sampled tokens
(replicated lm_head + same RNG counter) — is identical. So the only
coordination needed is broadcasting the request stream once per step:
one small broadcast_object_list on the TP group, no per-layer handshakes
beyond the weight all-reduces themselves.
"""

from __future__ import annotations

import logging
from typing import Any

import torch.distributed as dist

logger = logging.getLogger(__name__)


class TPLockstepEngine:
    """Wrap one TP rank's engine. The driver (tp rank 0) accepts requests
    and reads outputs; followers mirror the command stream in serve()."""

    def __init__(self, engine, tp_group=None, is_driver: bool | None = None):
        self.engine = engine
        self.tp_group = tp_group
        rank = dist.get_rank(group=tp_group) if dist.is_initialized() else 0
        self.is_driver = (rank == 0) if is_driver is None else is_driver
        self._pending: list[tuple[str, tuple, dict]] = []

    # -- driver-side request surface ----------------------------------------
    def add_request(self, *args, **kwargs) -> None:
        assert self.is_driver, "only the TP driver accepts requests"
        self._pending.append(("add_request", args, kwargs))

    def abort(self, request_id: str) -> None:
        assert self.is_driver
        self._pending.append(("abort", (request_id,), {}))

    def set_weight_version(self, v: int) -> None:
        assert self.is_driver
        self._pending.append(("weight_version", (v,), {}))

    # -- lockstep -----------------------------------------------------------
    def _exchange(self, cmd: dict | None) -> dict:
        if self.tp_group is None and not dist.is_initialized():
            return cmd or {"ops": [], "stop": False}
        box = [cmd]
        dist.broadcast_object_list(box, src=_group_src(self.tp_group), group=self.tp_group)
        return box[0]

    def _apply(self, ops: list) -> None:
        for name, args, kwargs in ops:
            if name == "weight_version":
                self.engine.weight_version = args[0]
            else:
                getattr(self.engine, name)(*args, **kwargs)

    def step(self) -> int:
        """Driver: flush pending commands to all ranks, then step. Returns
        engine.step()'s token count."""
        assert self.is_driver
        cmd = {"ops": self._pending, "stop": False}
        self._pending = []
        cmd = self._exchange(cmd)
        self._apply(cmd["ops"])
        return self.engine.step()

    def serve(self) -> None:
        """Follower loop: mirror the driver until stop."""
        assert not self.is_driver
        while True:
            cmd = self._exchange(None)
            if cmd.get("stop"):
                return
            self._apply(cmd["ops"])
            self.engine.step()

    def stop(self) -> None:
        assert self.is_driver
        self._exchange({"ops": [], "stop": True})

    # -- passthroughs (driver) ----------------------------------------------
    def has_unfinished(self) -> bool:
        return self.engine.has_unfinished()

    def pop_finished(self) -> list[Any]:
        return self.engine.pop_finished()

    def generate(self, prompts: list, params) -> list[Any]:
        """Blocking batch generation through the lockstep loop (bench/tests);
        mirrors LLMEngine.generate."""
        assert self.is_driver
        if not isinstance(params, (list, tuple)):
            params = [params] * len(prompts)
        for i, (p, sp) in enumerate(zip(prompts, params)):
            self.add_request(f"gen-{i}", p, sp)
        while True:
            self.step()
            if not self.engine.has_unfinished() and not self._pending:
                break
        self.stop()
        outs = {o.request_id: o for o in self.engine.pop_finished()}
        return [outs[f"gen-{i}"] for i in range(len(prompts))]


def _group_src(group) -> int:
    """Global rank of the group's rank-0 member."""
    if group is None:
        return 0
    return dist.get_global_rank(group, 0)
