#!/usr/bin/env python3
"""BASELINE config 1: single-turn GSM8K-style GRPO on the CPU plumbing
backend (no GPU needed). The full pipeline runs: gateway trace capture,
enrichment, GRPO advantages, policy update.

python examples/train_gsm8k_cpu.py
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import httpx

import rllm_amd
from rllm_amd.data.builders import synthetic_gsm8k
from rllm_amd.rewards.math_reward import math_reward_fn
from rllm_amd.trainer import AgentTrainer, TrainerConfig
from rllm_amd.trainer.algorithms.config import AlgorithmConfig


@rllm_amd.rollout
def solve(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user",
                                       "content": f"{task.instruction}\nAnswer with \\boxed{{}}."}]},
                   timeout=60.0)
    r.raise_for_status()
    return None


@rllm_amd.evaluator
def grade(task, episode):
    return math_reward_fn(task, episode)


def main():
    tasks = synthetic_gsm8k(16, seed=0).as_tasks(id_key="id")
    trainer = AgentTrainer(
        agent_flow=solve, evaluator=grade, train_dataset=tasks,
        backend="cpu",
        config=TrainerConfig(total_epochs=1, train_batch_size=4, rollout_n=4,
                             max_steps=2, logger_backends=["console"]),
        algorithm_config=AlgorithmConfig(estimator="grpo"),
        backend_kwargs={"rollout_max_tokens": 12},
    )
    trainer.train()


if __name__ == "__main__":
    main()
