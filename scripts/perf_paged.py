#!/usr/bin/env python3
"""paged_decode split-count sweep at the flagship decode shape
(B=256, Hk=2, seq 256..768): is the flash-decoding split (+merge kernel)
worth it once B*Hk already fills the chip?

  gpurun -- 'python scripts/perf_paged.py > gpurun_out/paged.log 2>&1'
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--seq", type=int, nargs="+", default=[256, 512, 768, 2048, 8192])
    ap.add_argument("--iters", type=int, default=200)
    args = ap.parse_args()

    from rllm_amd import ops

    Hq, Hk, D = 12, 2, 128
    PAGE = 16
    B = args.batch
    for S in args.seq:
        n_pages = (S + PAGE - 1) // PAGE
        total_pages = B * n_pages + 1
        k_pages = torch.randn(total_pages * PAGE, Hk, D, device="cuda").to(torch.bfloat16)
        v_pages = torch.randn(total_pages * PAGE, Hk, D, device="cuda").to(torch.bfloat16)
        q = torch.randn(B, Hq, D, device="cuda").to(torch.bfloat16)
        bt = torch.arange(1, B * n_pages + 1, device="cuda", dtype=torch.int32).reshape(B, n_pages)
        lens = torch.full((B,), S, device="cuda", dtype=torch.int32)
        scale = 1.0 / D ** 0.5

        line = [f"B={B} S={S}"]
        ref = None
        for splits in (1, 2, 4, 8, 0):  # 0 = heuristic
            o = ops.paged_decode(q, k_pages, v_pages, bt, lens, scale, n_splits=splits)
            if ref is None:
                ref = o.float()
            else:
                err = (o.float() - ref).abs().max().item()
                assert err < 0.02, f"splits={splits} diverged: {err}"
            torch.cuda.synchronize()
            t0 = time.monotonic()
            for _ in range(args.iters):
                ops.paged_decode(q, k_pages, v_pages, bt, lens, scale, n_splits=splits)
            torch.cuda.synchronize()
            us = (time.monotonic() - t0) / args.iters * 1e6
            line.append(f"s{splits}={us:7.1f}us")
        print("  ".join(line), flush=True)
        del k_pages, v_pages
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
