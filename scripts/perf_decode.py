#!/usr/bin/env python3
"""Decode-path microbenchmark: attributes per-step time to host-prep /
graph-replay / readback, and reports pure replay throughput.

Run on an MI355X: python scripts/perf_decode.py [--batch 64] [--steps 200]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--prompt-len", type=int, default=256)
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--no-graph", action="store_true")
    args = ap.parse_args()

    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel

    cfg = get_model_config(args.model)
    model = QwenModel(cfg, device="cuda").init_random(seed=0)
    engine = LLMEngine(model, kv_budget_bytes=32 << 30, eos_token_id=None,
                       use_hip_graph=not args.no_graph)

    warm = 32
    prompts = [list(range(100, 100 + args.prompt_len)) for _ in range(args.batch)]
    for i, p in enumerate(prompts):
        engine.add_request(f"s{i}", p, SamplingParams(temperature=1.0, max_tokens=warm + args.steps))
    # prefill
    while engine.waiting:
        engine.step()
    torch.cuda.synchronize()

    # warm the graph / block machinery
    while engine.running and min(len(s.output_ids) for s in engine.running) < warm:
        engine.step()
    torch.cuda.synchronize()

    # timed to completion (block decode: one step() call = up to 16 decode
    # positions via in-graph state advance, so count TOKENS, not calls;
    # raw graph.replay() loops are no longer valid — replays past the
    # reserved page block would walk off the block tables)
    tok_start = sum(len(s.output_ids) for s in engine.running)
    t0 = time.monotonic()
    while engine.has_unfinished():
        engine.step()
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    total_tokens = args.batch * (warm + args.steps)
    timed_tokens = total_tokens - tok_start
    per_pos = dt / (timed_tokens / args.batch)
    print(f"batch={args.batch} per_position={per_pos*1e3:.3f} ms  "
          f"decode_tok/s={timed_tokens/dt:,.0f}  (timed {timed_tokens} tokens)")

    # model-weight read lower bound
    bytes_weights = sum(p.numel() * p.element_size() for p in model.parameters())
    print(f"weights={bytes_weights/1e9:.2f} GB -> BW floor at 6.3TB/s = {bytes_weights/6.3e12*1e3:.3f} ms/step")


if __name__ == "__main__":
    main()
