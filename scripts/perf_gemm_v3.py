#!/usr/bin/env python3
"""COLD-weight decode-GEMM A/B: torch default vs tuned hipblaslt vs
skinny v2 vs skinny v3. Cycles weight copies past the 256 MB Infinity
Cache (hot-cache timing picks the wrong winners — see profiles/README)."""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from rllm_amd import ops

SHAPES = [
    ("qkv", 256, 2048, 1536),
    ("o", 256, 1536, 1536),
    ("gate_up", 256, 17920, 1536),
    ("down", 256, 1536, 8960),
    ("lm_head", 256, 151936, 1536),
]


def bench_cold(fn_for_w, wcopies, iters=40):
    for i in range(4):
        fn_for_w(wcopies[i % len(wcopies)])
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for i in range(iters):
        fn_for_w(wcopies[i % len(wcopies)])
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    C = ops.require_ext()
    print(f"{'shape':>8} {'N':>7} {'K':>5} {'torch':>8} {'tuned':>8} {'v2':>8} {'v3':>8} {'floor':>7}  (us, cold)")
    for name, M, N, K in SHAPES:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        n_cop = max(1, min(48, (384 << 20) // (N * K * 2) + 1))
        wc = [w] + [w.clone() for _ in range(n_cop - 1)]

        ref = torch.nn.functional.linear(a, w).float()
        out3 = C.skinny_gemm_v3(a, w, None).float()
        err = (out3 - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
        ok = "OK" if err < 0.05 else f"BAD({err:.3f})"

        ops.pretune_decode_shapes([(M, N, K)], verbose=False)
        t_torch = bench_cold(lambda wi: torch.nn.functional.linear(a, wi), wc)
        t_tuned = bench_cold(lambda wi: C.hbl_mm(a, wi), wc)
        t_v2 = bench_cold(lambda wi: C.skinny_gemm(a, wi, None), wc)
        t_v3 = bench_cold(lambda wi: C.skinny_gemm_v3(a, wi, None), wc)
        floor = (N * K * 2) / 6.3e12 * 1e6
        print(f"{name:>8} {N:>7} {K:>5} {t_torch:>8.1f} {t_tuned:>8.1f} {t_v2:>8.1f} {t_v3:>8.1f} {floor:>7.1f}  v3:{ok}")


if __name__ == "__main__":
    main()
