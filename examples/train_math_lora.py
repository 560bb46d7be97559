#!/usr/bin/env python3
"""LoRA GRPO on MATH-style tasks: adapter-only optimizer state and
KL against the adapter-disabled base (no second model copy).

python examples/train_math_lora.py [--model r1-distill-qwen-1.5b] [--steps 2] [--rank 16]
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import httpx

import rllm_amd
from rllm_amd.data.builders import synthetic_gsm8k
from rllm_amd.models.lora import LoRAConfig
from rllm_amd.rewards.math_reward import math_reward_fn
from rllm_amd.trainer import TrainerConfig
from rllm_amd.trainer.agent_trainer import AgentTrainer
from rllm_amd.trainer.policy import PolicyTrainerConfig


@rllm_amd.rollout
def solve(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user",
                                       "content": f"{task.instruction}\nAnswer with \\boxed{{}}."}],
                         "max_tokens": 256},
                   timeout=600.0)
    r.raise_for_status()


@rllm_amd.evaluator
def grade(task, episode):
    return math_reward_fn(task, episode)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--rank", type=int, default=16)
    args = ap.parse_args()

    tasks = synthetic_gsm8k(n=64, seed=0).as_tasks(id_key="id")
    trainer = AgentTrainer(
        agent_flow=solve,
        evaluator=grade,
        train_dataset=tasks,
        backend="native",
        backend_kwargs=dict(
            model_config=args.model,
            lora=LoRAConfig(r=args.rank, alpha=2 * args.rank),
            # kl_beta > 0 + LoRA => KL computed against the frozen base
            # (adapters disabled), not a separate reference model
            policy_config=PolicyTrainerConfig(lr=1e-4, kl_beta=1e-3),
            rollout_sampling_params={"temperature": 1.0, "max_tokens": 256},
        ),
        config=TrainerConfig(train_batch_size=16, rollout_n=8,
                             max_steps=args.steps, logger_backends=["console"]),
    )
    trainer.train()


if __name__ == "__main__":
    main()
