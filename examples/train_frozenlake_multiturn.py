#!/usr/bin/env python3
"""BASELINE config 4 shape: multi-turn FrozenLake agent with RLOO and the
gateway's cumulative token mode (each turn's prompt extends the previous
turn's prompt+completion token-for-token, so the trainer packs one merged
row per episode with response_mask = [0s 1s 0s 1s ...]).

python examples/train_frozenlake_multiturn.py [--steps 2] [--model tiny]
"""

import argparse
import re
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import httpx

import rllm_amd
from rllm_amd.environments.frozenlake import ACTIONS, FrozenLakeEnv
from rllm_amd.gateway.models import GatewayConfig
from rllm_amd.trainer import TrainerConfig
from rllm_amd.trainer.agent_trainer import AgentTrainer
from rllm_amd.trainer.algorithms.config import AlgorithmConfig
from rllm_amd.trainer.policy import PolicyTrainerConfig

MAX_TURNS = 4


def _parse_action(reply: str) -> str:
    m = re.search(r"\b(left|down|right|up)\b", reply.lower())
    return m.group(1) if m else "noop"


def _replay(episode) -> float:
    """Deterministic env replay over the episode's recorded actions."""
    env = FrozenLakeEnv()
    env.reset()
    reward, done = 0.0, False
    for traj in episode.trajectories:
        for step in traj.steps:
            if done:
                break
            _, reward, done, _ = env.step(_parse_action(step.model_response or ""))
    return float(reward)


@rllm_amd.rollout
def play(task, config):
    env = FrozenLakeEnv()
    obs, _ = env.reset()
    msgs = [{"role": "user",
             "content": "You are P on a frozen lake (F frozen, H hole, G goal). "
                        f"Reach G.\n{obs}\nReply with one of: {', '.join(ACTIONS)}."}]
    for _ in range(MAX_TURNS):
        r = httpx.post(config.base_url + "/chat/completions",
                       json={"model": config.model, "messages": msgs, "max_tokens": 8},
                       timeout=600.0)
        r.raise_for_status()
        reply = r.json()["choices"][0]["message"]["content"]
        obs, reward, done, _ = env.step(_parse_action(reply))
        if done:
            break
        msgs = msgs + [{"role": "assistant", "content": reply},
                       {"role": "user", "content": f"{obs}\nNext move?"}]
    return None


@rllm_amd.evaluator
def reached_goal(task, episode):
    return _replay(episode)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--model", default="tiny")
    args = ap.parse_args()

    tasks = [{"id": str(i), "question": f"episode {i}"} for i in range(8)]

    trainer = AgentTrainer(
        agent_flow=play,
        evaluator=reached_goal,
        train_dataset=tasks,
        backend="native",
        backend_kwargs=dict(
            model_config=args.model,
            policy_config=PolicyTrainerConfig(lr=1e-5, kl_beta=0.0),
            rollout_sampling_params={"temperature": 1.0, "max_tokens": 8},
            gateway_config=GatewayConfig(cumulative_token_mode=True),
        ),
        config=TrainerConfig(train_batch_size=4, rollout_n=4,
                             max_steps=args.steps, logger_backends=["console"]),
        algorithm_config=AlgorithmConfig(estimator="rloo"),
    )
    trainer.train()


if __name__ == "__main__":
    main()
