"""Step-gated GPU profiling hooks (reference: trainer.profile_steps →
start/stop_profiling worker RPCs + Nsight options, verl_backend.py:853-870.
MI355X equivalent: roctx ranges + the HIP profiler start/stop API, which
rocprofv3 consumes:

    rocprofv3 --kernel-trace --stats --output-format csv \
        -d out -- python train.py   # ranges mark the chosen steps

torch.cuda.profiler.start()/stop() maps to hipProfilerStart/Stop and
torch.cuda.nvtx maps to roctx on ROCm builds — no CUDA shim involved."""

from __future__ import annotations

from contextlib import contextmanager

import torch


@contextmanager
def step_profile_region(step: int, profile_steps=None, tag: str = "train_step"):
    """Wrap one training step; activates the GPU profiler only for steps
    listed in `profile_steps` (no-op otherwise and on CPU)."""
    active = bool(profile_steps) and step in profile_steps and torch.cuda.is_available()
    if not active:
        yield False
        return
    torch.cuda.synchronize()
    torch.cuda.profiler.start()
    torch.cuda.nvtx.range_push(f"{tag}_{step}")
    try:
        yield True
    finally:
        torch.cuda.nvtx.range_pop()
        torch.cuda.synchronize()
        torch.cuda.profiler.stop()


@contextmanager
def roctx_range(name: str):
    """Named roctx range (visible in rocprofv3 marker traces)."""
    if torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
