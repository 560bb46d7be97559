#!/usr/bin/env python3
"""Separated-placement GRPO: trainer ranks + dedicated rollout ranks on one
node (parallel/separated.py topology; the verl separated-placement role).

    torchrun --nnodes 1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        scripts/run_separated.py --trainers 4 --steps 3

Ranks [0, T) train (DP over RCCL); ranks [T, world) each run a full
rollout engine replica. Trainer 0 drives the fleet: one flat bf16 weight
broadcast per version, task dispatch via the command loop, results
gathered back. Use when rollout and update want different GPU counts
(e.g. long rollouts, short updates) — the colocated default (bench.py)
remains the better fit when both phases saturate the same GPUs.
"""

from __future__ import annotations

import argparse
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trainers", type=int, default=4)
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--tasks", type=int, default=32, help="tasks per step (global)")
    ap.add_argument("--rollout-n", type=int, default=8)
    ap.add_argument("--prompt-len", type=int, default=256)
    ap.add_argument("--max-new-tokens", type=int, default=512)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.parallel import dist as pdist
    from rllm_amd.parallel.separated import (RolloutWorker, SeparatedRolloutClient,
                                             SeparatedTopology)

    rank, world, local = pdist.init_from_env()
    torch.cuda.set_device(f"cuda:{local}")
    topo = SeparatedTopology(n_trainers=args.trainers, n_rollout=world - args.trainers)
    # scope ALL pdist collectives (PolicyTrainer's grad all-reduce, token
    # denominators, object gathers) to the trainer subgroup — rollout ranks
    # live in their own command loop
    pdist.set_default_group(topo.trainer_group)
    cfg = get_model_config(args.model)
    model = QwenModel(cfg, device=f"cuda:{local}").init_random(seed=args.seed)

    if topo.is_rollout:
        # repoint the model's params into ONE flat buffer BEFORE the engine
        # captures any graph: the weight broadcast then updates the live
        # weights in place (torch.cat would broadcast into a dead copy)
        from rllm_amd.trainer.optim import flatten_params

        flat, _ = flatten_params(model)
        engine = LLMEngine(model, kv_budget_bytes=64 << 30, eos_token_id=None,
                           seed=args.seed + rank)
        sp = SamplingParams(temperature=1.0, max_tokens=args.max_new_tokens)

        def generate(prompt_lists):
            outs = engine.generate(prompt_lists, sp)
            return [{"prompt_ids": o.prompt_ids, "token_ids": o.token_ids,
                     "logprobs": o.logprobs} for o in outs]

        RolloutWorker(topo, flat, generate, engine=engine).serve()
        pdist.destroy()
        return

    # ---- trainer ranks ----
    from rllm_amd.trainer.batch import PackedRow
    from rllm_amd.trainer.policy import PolicyTrainer, PolicyTrainerConfig

    trainer = PolicyTrainer(model, None, PolicyTrainerConfig(lr=1e-6, kl_beta=0.0,
                                                             use_ref=False,
                                                             old_logprob_mode="rollout"))
    client = SeparatedRolloutClient(topo, trainer.flat_param)
    rng = np.random.default_rng(args.seed)

    for step in range(args.steps):
        t0 = time.monotonic()
        results = None
        if topo.rank == 0:
            client.sync_weights(version=step + 1)
            prompts = [rng.integers(0, cfg.vocab_size, size=args.prompt_len).tolist()
                       for _ in range(args.tasks * args.rollout_n)]
            results = client.generate(prompts)
        # fan results out to the other trainer ranks for DP update
        results = pdist.all_gather_object_list(results)[0]
        mine = results[topo.rank :: topo.n_trainers]
        rows = []
        for i, r in enumerate(mine):
            toks = r["prompt_ids"] + r["token_ids"]
            mask = [0] * len(r["prompt_ids"]) + [1] * len(r["token_ids"])
            adv = float(rng.standard_normal())
            rows.append(PackedRow(tokens=toks, response_mask=mask,
                                  advantages=[adv] * len(toks),
                                  rollout_logprobs=[0.0] * len(r["prompt_ids"]) + r["logprobs"]))
        metrics = trainer.update_policy(rows)
        if topo.rank == 0:
            dt = time.monotonic() - t0
            toks = sum(len(r["token_ids"]) for r in results)
            print(f"[step {step}] {toks/dt:,.0f} tok/s  loss={metrics.get('actor/loss'):.4f}  "
                  f"({dt:.2f}s, {topo.n_trainers} trainers + {topo.n_rollout} rollout)")

    if topo.rank == 0:
        client.stop()
    pdist.destroy()


if __name__ == "__main__":
    main()
