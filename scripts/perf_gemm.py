#!/usr/bin/env python3
"""Decode-GEMM shape microbench: hipBLASLt (torch.matmul / F.linear) vs the
in-repo skinny_gemm kernel, with the weight-stream bandwidth floor."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from rllm_amd import ops

SHAPES = [  # (name, M, N, K)
    ("qkv", 256, 2304, 1536),
    ("o", 256, 1536, 1536),
    ("gate_up", 256, 17920, 1536),
    ("down", 256, 1536, 8960),
    ("lm_head", 256, 151936, 1536),
]


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def main():
    torch.manual_seed(0)
    print(f"{'shape':>8} {'M':>4} {'N':>7} {'K':>5} {'blaslt_us':>10} {'skinny_us':>10} {'floor_us':>9}")
    for name, M, N, K in SHAPES:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        t_blas = bench(lambda: torch.nn.functional.linear(a, w))
        t_skinny = bench(lambda: ops.skinny_gemm(a, w, None))
        floor = (N * K * 2) / 6.3e12
        print(f"{name:>8} {M:>4} {N:>7} {K:>5} {t_blas*1e6:>10.1f} {t_skinny*1e6:>10.1f} {floor*1e6:>9.1f}")


if __name__ == "__main__":
    main()
