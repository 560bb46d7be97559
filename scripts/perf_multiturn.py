#!/usr/bin/env python3
"""Multi-turn rollout benchmark: cumulative sessions (turn N+1's prompt =
turn N's prompt + completion + a small delta) with prefix caching ON vs
OFF. This is the config-4 agent shape where APC pays — later turns
prefill only the delta instead of the whole accumulated context.

python scripts/perf_multiturn.py [--sessions 64] [--turns 3]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch


def run(model, *, sessions, turns, prompt_len, new_tokens, delta_len, cache):
    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams

    engine = LLMEngine(model, kv_budget_bytes=48 << 30, eos_token_id=None,
                       seed=3, enable_prefix_caching=cache, max_model_len=16384)
    g = torch.Generator().manual_seed(9)
    prompts = [torch.randint(10, 10000, (prompt_len,), generator=g).tolist()
               for _ in range(sessions)]

    torch.cuda.synchronize()
    t0 = time.monotonic()
    total_new = 0
    histories = prompts
    for turn in range(turns):
        outs = engine.generate(histories, SamplingParams(temperature=1.0, max_tokens=new_tokens))
        total_new += sum(len(o.token_ids) for o in outs)
        delta = torch.randint(10, 10000, (delta_len,), generator=g).tolist()
        histories = [h + o.token_ids + delta for h, o in zip(histories, outs)]
    torch.cuda.synchronize()
    dt = time.monotonic() - t0
    hits = engine.prefix_cache.hits if engine.prefix_cache else 0
    del engine
    torch.cuda.empty_cache()
    return dt, total_new, hits


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sessions", type=int, default=64)
    ap.add_argument("--turns", type=int, default=3)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--new-tokens", type=int, default=128)
    ap.add_argument("--delta-len", type=int, default=32)
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    args = ap.parse_args()

    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel

    model = QwenModel(get_model_config(args.model), device="cuda").init_random(seed=0)
    kw = dict(sessions=args.sessions, turns=args.turns, prompt_len=args.prompt_len,
              new_tokens=args.new_tokens, delta_len=args.delta_len)

    dt_off, new_off, _ = run(model, cache=False, **kw)
    dt_on, new_on, hits = run(model, cache=True, **kw)
    assert new_off == new_on
    print(f"sessions={args.sessions} turns={args.turns} prompt={args.prompt_len} "
          f"new/turn={args.new_tokens} delta={args.delta_len}")
    print(f"cache OFF: {dt_off:.2f}s  ({new_off/dt_off:,.0f} new tok/s)")
    print(f"cache ON:  {dt_on:.2f}s  ({new_on/dt_on:,.0f} new tok/s)  "
          f"hits={hits} pages  speedup={dt_off/dt_on:.2f}x")


if __name__ == "__main__":
    main()
