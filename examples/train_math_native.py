#!/usr/bin/env python3
"""BASELINE config 2 shape: GRPO on MATH-style tasks with the native
MI355X backend (DeepSeek-R1-Distill-Qwen-1.5B architecture, random init —
no network for checkpoints; pass --checkpoint + --tokenizer for real
weights).

python examples/train_math_native.py [--model r1-distill-qwen-1.5b] [--steps 2]
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import httpx

import rllm_amd
from rllm_amd.data.builders import synthetic_gsm8k
from rllm_amd.rewards.math_reward import math_reward_fn
from rllm_amd.trainer import AgentTrainer, TrainerConfig
from rllm_amd.trainer.algorithms.config import AlgorithmConfig
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
    return None


@rllm_amd.evaluator
def grade(task, episode):
    return math_reward_fn(task, episode)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--tokenizer", default=None, help="local HF tokenizer dir")
    ap.add_argument("--checkpoint", default=None, help="actor state dict (.pt)")
    args = ap.parse_args()

    from rllm_amd.utils.tokenizer import load_tokenizer

    tasks = synthetic_gsm8k(32, seed=0).as_tasks(id_key="id")
    trainer = AgentTrainer(
        agent_flow=solve, evaluator=grade, train_dataset=tasks,
        backend="native",
        config=TrainerConfig(total_epochs=1, train_batch_size=8, rollout_n=8,
                             max_steps=args.steps, logger_backends=["console"]),
        algorithm_config=AlgorithmConfig(estimator="grpo", kl_beta=1e-3),
        backend_kwargs={
            "model_config": args.model,
            "tokenizer": load_tokenizer(args.tokenizer),
            "checkpoint_path": args.checkpoint,
            "policy_config": PolicyTrainerConfig(lr=1e-6, kl_beta=1e-3),
            "rollout_sampling_params": {"temperature": 1.0, "top_p": 1.0, "max_tokens": 256},
        },
    )
    trainer.train()


if __name__ == "__main__":
    main()
