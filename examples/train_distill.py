#!/usr/bin/env python3
"""On-policy distillation: the student rolls out, a teacher served on any
OpenAI-compatible endpoint scores the realized tokens (echo-mode
completions logprobs), and per-token reverse-KL advantages drive the
update through the precomputed-advantage estimator (reference
agent_workflow_trainer.py:704-766).

python examples/train_distill.py --teacher-url http://teacher-host:8000/v1 \
    --teacher-model big-model [--model r1-distill-qwen-1.5b]
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import rllm_amd
from rllm_amd.data.builders import synthetic_gsm8k
from rllm_amd.trainer.algorithms.config import AlgorithmConfig
from rllm_amd.trainer.distill import TeacherClient
from rllm_amd.trainer.native_backend import NativeBackend
from rllm_amd.trainer.policy import PolicyTrainerConfig
from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer


@rllm_amd.rollout
def solve(task, config):
    r = config.post("/chat/completions",
                    json={"model": config.model,
                          "messages": [{"role": "user", "content": str(task.instruction)}],
                          "max_tokens": 256})
    r.raise_for_status()
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--teacher-url", required=True)
    ap.add_argument("--teacher-model", required=True)
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--coef", type=float, default=1.0)
    ap.add_argument("--clip", type=float, default=5.0)
    args = ap.parse_args()

    tasks = synthetic_gsm8k(n=128)
    backend = NativeBackend(
        solve, model_config=args.model,
        # distillation needs no clip/KL-to-ref: the teacher KL IS the signal
        policy_config=PolicyTrainerConfig(lr=1e-6, kl_beta=0.0, grad_clip=1.0,
                                          old_logprob_mode="alias", use_ref=False),
        rollout_sampling_params={"temperature": 1.0, "max_tokens": 256},
        distill={"teacher": TeacherClient(args.teacher_url, args.teacher_model),
                 "coef": args.coef, "clip": args.clip})

    trainer = UnifiedTrainer(
        backend, tasks,
        config=TrainerConfig(train_batch_size=16, rollout_n=4, max_steps=args.steps),
        algorithm_config=AlgorithmConfig(use_precomputed_advantage=True))
    trainer.fit()


if __name__ == "__main__":
    main()
