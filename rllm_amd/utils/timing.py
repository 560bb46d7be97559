"""Timing helpers (reference: verl-style simple_timer/marked_timer feeding
timing_dict → throughput metrics)."""

from __future__ import annotations

import time
from contextlib import contextmanager


@contextmanager
def simple_timer(name: str, timing_dict: dict):
    t0 = time.monotonic()
    try:
        yield
    finally:
        timing_dict[name] = timing_dict.get(name, 0.0) + time.monotonic() - t0


@contextmanager
def marked_timer(name: str, timing_dict: dict, *, color: str | None = None):
    """Same as simple_timer; the marker arg mirrors the reference's
    profiler-range API (rocprofv3 ranges can hook here later)."""
    with simple_timer(name, timing_dict):
        yield


def compute_throughput_metrics(timing_dict: dict, n_tokens: int, n_gpus: int = 1) -> dict:
    """tokens/sec whole-job + per-GPU from a step timing dict
    (reference verl_backend.py:884-889)."""
    total = sum(v for k, v in timing_dict.items() if k.startswith("time/")) or \
        timing_dict.get("time/step_s", 0.0) or sum(timing_dict.values())
    if total <= 0:
        return {}
    return {
        "perf/throughput_tokens_per_s": n_tokens / total,
        "perf/throughput_tokens_per_s_per_gpu": n_tokens / total / max(1, n_gpus),
        "perf/total_time_s": total,
    }


def interval_union(intervals: list[tuple[float, float]]) -> float:
    """Total covered wall time of possibly-overlapping [start, end) spans
    (reference agentflow_engine.py:251-320 LLM wall-time summary)."""
    if not intervals:
        return 0.0
    spans = sorted(intervals)
    total = 0.0
    cur_s, cur_e = spans[0]
    for s, e in spans[1:]:
        if s > cur_e:
            total += cur_e - cur_s
            cur_s, cur_e = s, e
        else:
            cur_e = max(cur_e, e)
    total += cur_e - cur_s
    return total
