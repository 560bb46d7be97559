#!/usr/bin/env python3
"""Flagship benchmark: GRPO train tokens/sec (rollout+update) for
DeepSeek-R1-Distill-Qwen-1.5B on MATH-shaped synthetic data (BASELINE.json
metric), 1..8 MI355X, one rank per GPU over RCCL.

One step = the reference's 8-stage on-policy batch:
  S1 rollout: n=8 samples/task via the continuous-batching engine
     (flash prefill + paged decode HIP kernels, fused sampling+logprob)
  S2-S3 transform episodes -> trajectory groups (GRPO grouping)
  S5 old-logprob recompute + frozen-ref logprob (chunked fused logprob)
  S6 GRPO advantages (numpy, per group)
  S7 update: fused ratio-clip+KL loss, backward, RCCL grad all-reduce,
     fused AdamW
Algorithmic config mirrors the reference defaults: rollout.n=8, temp 1.0,
top_p 1.0, clip 0.2, KL-in-loss low_var_kl coef 1e-3, lr 1e-6, grad_clip
1.0, token-mean loss agg (BASELINE.md / base.yaml:49-60, yaml:9-43).

Usage: python bench.py --gpus N --steps K --warmup W
(launched by the driver via torch.distributed.run for N>1; reads
RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env).
"""

from __future__ import annotations

import argparse
import copy
import json
import os
import time

import numpy as np
import torch

WALL_T0 = time.monotonic()


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", type=str, default="r1-distill-qwen-1.5b")
    p.add_argument("--tasks-per-gpu", type=int, default=32, help="tasks per rank per step (weak scaling; 32x8=256 seqs = reference ppo_mini_batch_size)")
    p.add_argument("--rollout-n", type=int, default=8, help="samples per task (reference default 8)")
    p.add_argument("--prompt-len", type=int, default=256, help="synthetic MATH-shaped prompt length")
    p.add_argument("--max-new-tokens", type=int, default=512, help="response length cap per rollout")
    p.add_argument("--micro-tokens", type=int, default=0,
                   help="ppo_max_token_len_per_gpu (0 = auto by model size)")
    p.add_argument("--max-batched-tokens", type=int, default=8192,
                   help="prefill chunk size (engine knob, not an algorithmic config)")
    p.add_argument("--kl-beta", type=float, default=1e-3)
    p.add_argument("--lr", type=float, default=1e-6)
    p.add_argument("--old-logprob-mode", default="alias",
                   choices=["recompute", "alias", "rollout"],
                   help="'alias' (default) is BIT-identical to 'recompute' at one optimizer "
                        "step per batch (proven by tests/test_engine_gpu.py::"
                        "test_alias_old_logprob_bit_identical_to_recompute) but skips the "
                        "redundant no-grad forward")
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument("--dist-backend", default=None, choices=[None, "nccl", "gloo"],
                   help="override the collective backend (gloo lets a 2-rank "
                        "smoke run share ONE GPU, which RCCL forbids)")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    assert torch.cuda.is_available(), "bench.py requires an MI355X (HIP) device"

    from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams
    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.parallel import dist as pdist
    from rllm_amd.trainer.algorithms.advantage import collect_reward_and_advantage_from_trajectory_groups
    from rllm_amd.trainer.algorithms.config import AlgorithmConfig, TransformConfig
    from rllm_amd.trainer.algorithms.transform import transform_episodes_to_trajectory_groups
    from rllm_amd.trainer.batch import rows_from_groups
    from rllm_amd.trainer.policy import PolicyTrainer, PolicyTrainerConfig
    from rllm_amd.types import Episode, Step, Trajectory

    pdist.init_from_env(backend=args.dist_backend)
    # gloo smoke mode: ranks share GPU 0 (a 1-GPU box); RCCL mode: one GPU
    # per rank as torchrun assigns
    n_devices = torch.cuda.device_count()
    device = f"cuda:{local_rank % n_devices}"
    torch.cuda.set_device(device)

    cfg = get_model_config(args.model)
    torch.manual_seed(args.seed + rank)

    if args.micro_tokens <= 0:
        # activation memory per token scales with hidden+intermediate width;
        # keep saved activations well under HBM alongside actor+ref+optim+KV
        n_params_b = cfg.param_count() / 1e9
        args.micro_tokens = 32768 if n_params_b < 3 else (8192 if n_params_b < 10 else 4096)

    # Actor (random init — no network for checkpoints) + frozen reference.
    # 288 GB HBM: both co-resident; rollout engine SHARES the actor weights
    # (colocated weight sync = zero-copy).
    model = QwenModel(cfg, device=device).init_random(seed=args.seed)
    ref_model = QwenModel(cfg, device=device).init_random(seed=args.seed)
    for p in ref_model.parameters():
        p.requires_grad_(False)

    trainer = PolicyTrainer(
        model, ref_model,
        PolicyTrainerConfig(
            lr=args.lr, kl_beta=args.kl_beta, eps_clip=0.2, grad_clip=1.0,
            max_tokens_per_micro=args.micro_tokens, loss_agg_mode="token-mean",
            use_ref=args.kl_beta > 0, old_logprob_mode=args.old_logprob_mode,
        ))

    free, _ = torch.cuda.mem_get_info()
    # ranks sharing one device (gloo smoke on a 1-GPU box) split the budget
    ranks_per_dev = max(1, (world + n_devices - 1) // n_devices)
    engine = LLMEngine(
        model, max_num_seqs=1024, max_num_batched_tokens=args.max_batched_tokens,
        kv_budget_bytes=min(int(free * 0.35) // ranks_per_dev, 64 << 30),
        eos_token_id=None,  # synthetic data: length-capped rollouts
        seed=args.seed * 1000 + rank)

    n_seqs = args.tasks_per_gpu * args.rollout_n
    rng = np.random.default_rng(args.seed + 17 * rank)
    sp = SamplingParams(temperature=1.0, top_p=1.0, max_tokens=args.max_new_tokens)

    phase_times = {"rollout_s": 0.0, "update_s": 0.0}

    def one_step(step_idx: int) -> int:
        """Returns number of response tokens generated on this rank."""
        # --- S1 rollout ---
        t_r0 = time.monotonic()
        prompts = [rng.integers(0, cfg.vocab_size, size=args.prompt_len).tolist() for _ in range(n_seqs)]
        outs = engine.generate(prompts, sp)
        torch.cuda.synchronize()
        phase_times["rollout_s"] += time.monotonic() - t_r0
        t_u0 = time.monotonic()

        # --- episodes with synthetic rule rewards (group-variant) ---
        episodes = []
        for t in range(args.tasks_per_gpu):
            for ri in range(args.rollout_n):
                o = outs[t * args.rollout_n + ri]
                reward = float(rng.random() < 0.5)
                step = Step(prompt_ids=o.prompt_ids, response_ids=o.token_ids,
                            logprobs=o.logprobs, chat_completions=[{"role": "user", "content": "m"}],
                            reward=reward, done=True)
                traj = Trajectory(name="solver", steps=[step], reward=reward)
                episodes.append(Episode(id=f"r{rank}t{t}:{ri}", task={"i": t},
                                        trajectories=[traj], is_correct=reward > 0))

        # --- S2 transform + S6 advantages (GRPO per group) ---
        groups, _ = transform_episodes_to_trajectory_groups(episodes, TransformConfig())
        collect_reward_and_advantage_from_trajectory_groups(groups, AlgorithmConfig())
        rows = rows_from_groups(groups)

        # --- S5 old/ref logprobs + S7 update (fused kernels) ---
        def old_lp_fn(batch, rows_idx):
            with torch.no_grad():
                hidden = model.forward_train(batch.input_ids, batch.positions, batch.cu_seqlens)
                from rllm_amd import ops
                lp, _ = ops.chunked_logprob(hidden[rows_idx], model.lm_weight,
                                            batch.targets[rows_idx], want_entropy=False)
                return lp
        trainer.update_policy(rows, old_logprob_fn=old_lp_fn)
        engine.weight_version = trainer.weight_version  # colocated sync: shared tensors
        torch.cuda.synchronize()
        phase_times["update_s"] += time.monotonic() - t_u0
        return sum(len(o.token_ids) for o in outs)

    # --- warmup ---
    for w in range(args.warmup):
        one_step(-1 - w)
    phase_times["rollout_s"] = 0.0
    phase_times["update_s"] = 0.0

    # --- timed region ---
    pdist.barrier()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    resp_tokens = 0
    for s in range(args.steps):
        resp_tokens += one_step(s)
    torch.cuda.synchronize()
    pdist.barrier()
    t1 = time.monotonic()

    elapsed = t1 - t0
    elapsed = pdist.all_reduce_scalar(elapsed, op="max")  # max over ranks
    total_tokens = pdist.all_reduce_scalar(float(resp_tokens), op="sum")
    tokens_per_s = total_tokens / elapsed

    if rank == 0:
        result = {
            "metric": "GRPO train tokens/sec (rollout+update), 1.5B on MATH",
            "value": round(tokens_per_s, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": ("DeepSeek-R1-Distill-Qwen-1.5B (random init)" if args.model == "r1-distill-qwen-1.5b" else f"{args.model} (random init)"),
                "global_batch": n_seqs * n_gpus,
                "seq_len": args.prompt_len + args.max_new_tokens,
                "parallelism": f"dp{n_gpus}",
                "rollout_n": args.rollout_n,
                "prompt_len": args.prompt_len,
                "max_new_tokens": args.max_new_tokens,
                "temperature": 1.0, "top_p": 1.0, "clip_ratio": 0.2,
                "kl_beta": args.kl_beta, "lr": args.lr,
                "loss_agg": "token-mean",
                "old_logprob_mode": args.old_logprob_mode + (
                    " (bit-identical to recompute at 1 optim step/batch; "
                    "test_alias_old_logprob_bit_identical_to_recompute)"
                    if args.old_logprob_mode == "alias" else ""),
                "rollout_ms_per_step": round(phase_times["rollout_s"] / args.steps * 1000, 1),
                "update_ms_per_step": round(phase_times["update_s"] / args.steps * 1000, 1),
                "note": "tokens = response tokens generated AND trained on per wall-clock second, whole job",
            },
        }
        print(json.dumps(result))

    pdist.destroy()


if __name__ == "__main__":
    main()
