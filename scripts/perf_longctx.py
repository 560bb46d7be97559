#!/usr/bin/env python3
"""Long-context capability measurement (VERDICT r2 #3 — the DeepScaleR
shape: 8K/16K/24K response training on one MI355X, reference
docs/projects/deep-scaler.mdx:20).

Per response length: one full GRPO step (rollout via the continuous-
batching engine + alias-mode update) over `--seqs` sequences with a 1K
prompt. Reports rollout/update wall time, tokens/s, and HBM headroom.

  gpurun -- 'python scripts/perf_longctx.py --lens 8192 16384 24576 \
             --seqs 16 > gpurun_out/longctx.log 2>&1'
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--lens", type=int, nargs="+", default=[8192, 16384, 24576])
    ap.add_argument("--seqs", type=int, default=16)
    ap.add_argument("--prompt-len", type=int, default=1024)
    ap.add_argument("--micro-tokens", type=int, default=32768)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.trainer.batch import PackedRow
    from rllm_amd.trainer.policy import PolicyTrainer, PolicyTrainerConfig

    cfg = get_model_config(args.model)
    max_len = args.prompt_len + max(args.lens) + 64
    model = QwenModel(cfg, device="cuda").init_random(seed=args.seed)
    ref = QwenModel(cfg, device="cuda").init_random(seed=args.seed)
    for p in ref.parameters():
        p.requires_grad_(False)
    trainer = PolicyTrainer(model, ref, PolicyTrainerConfig(
        lr=1e-6, kl_beta=1e-3, max_tokens_per_micro=args.micro_tokens,
        old_logprob_mode="alias"))

    free, total = torch.cuda.mem_get_info()
    engine = LLMEngine(model, max_num_seqs=256, max_num_batched_tokens=8192,
                       kv_budget_bytes=min(int(free * 0.5), 96 << 30),
                       eos_token_id=None, seed=7, max_model_len=max_len)
    rng = np.random.default_rng(args.seed)

    for L in args.lens:
        prompts = [rng.integers(0, cfg.vocab_size, size=args.prompt_len).tolist()
                   for _ in range(args.seqs)]
        sp = SamplingParams(temperature=1.0, top_p=1.0, max_tokens=L)
        torch.cuda.synchronize()
        t0 = time.monotonic()
        outs = engine.generate(prompts, sp)
        torch.cuda.synchronize()
        t_roll = time.monotonic() - t0
        gen = sum(len(o.token_ids) for o in outs)

        rows = []
        for o in outs:
            toks = o.prompt_ids + o.token_ids
            mask = [0] * len(o.prompt_ids) + [1] * len(o.token_ids)
            rows.append(PackedRow(tokens=toks, response_mask=mask,
                                  advantages=[0.5] * len(toks),
                                  rollout_logprobs=[0.0] * len(o.prompt_ids) + o.logprobs))
        torch.cuda.synchronize()
        t0 = time.monotonic()
        trainer.update_policy(rows)
        torch.cuda.synchronize()
        t_upd = time.monotonic() - t0

        free2, _ = torch.cuda.mem_get_info()
        alloc = torch.cuda.memory_allocated() / (1 << 30)
        peak = torch.cuda.max_memory_allocated() / (1 << 30)
        tps = gen / (t_roll + t_upd)
        print(f"L={L:6d} seqs={args.seqs:3d} gen_tokens={gen:8d} "
              f"rollout={t_roll:7.2f}s update={t_upd:6.2f}s "
              f"train_tok/s={tps:9.1f} decode_tok/s={gen / t_roll:9.1f} "
              f"alloc={alloc:6.1f}GB peak={peak:6.1f}GB free={free2 / (1 << 30):6.1f}GB",
              flush=True)
        torch.cuda.reset_peak_memory_stats()


if __name__ == "__main__":
    main()
