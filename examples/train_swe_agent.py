#!/usr/bin/env python3
"""BASELINE config 5 shape: train a CLI coding agent (mini-swe-agent by
default) on SWE-bench-style tasks with sandboxed verifier rewards.

The dataset is a Harbor task-per-directory tree (build one offline from
SWE-bench rows with `rllm_amd.data.swe_builders.build_swebench_dataset`,
or point --dataset-dir at an existing one). Each task's docker image is
pulled by the sandbox backend at rollout time; the verifier
(tests/test.sh) runs IN the sandbox and its exit code is the reward.

python examples/train_swe_agent.py --dataset-dir ./swebench_tasks \
    [--harness mini-swe-agent] [--model-dir /path/to/hf/qwen2.5-7b]
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from rllm_amd.harnesses import get_harness
from rllm_amd.hooks import SandboxTaskHooks
from rllm_amd.sandbox.backends import DockerSandboxBackend
from rllm_amd.sandbox.manager import SandboxManager
from rllm_amd.tasks.loader import load_tasks
from rllm_amd.trainer.native_backend import NativeBackend
from rllm_amd.trainer.policy import PolicyTrainerConfig
from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset-dir", required=True)
    ap.add_argument("--harness", default="mini-swe-agent")
    ap.add_argument("--model-dir", default="r1-distill-qwen-1.5b",
                    help="registry name or local HF model dir")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=8)
    ap.add_argument("--rollout-n", type=int, default=4)
    ap.add_argument("--max-tokens", type=int, default=4096)
    ap.add_argument("--n-parallel", type=int, default=32)
    args = ap.parse_args()

    tasks = load_tasks(args.dataset_dir)
    print(f"{len(tasks)} tasks from {args.dataset_dir}")

    # the harness IS the agent flow; the sandbox hooks resolve each task's
    # verifier (tests/test.sh) into a sandbox-bound evaluator
    harness = get_harness(args.harness)
    hooks = SandboxTaskHooks(sandbox_manager=SandboxManager(DockerSandboxBackend()))

    backend = NativeBackend(
        harness,
        model_config=args.model_dir,
        policy_config=PolicyTrainerConfig(
            lr=1e-6, kl_beta=1e-3, grad_clip=1.0, old_logprob_mode="alias"),
        rollout_sampling_params={"temperature": 1.0, "top_p": 1.0,
                                 "max_tokens": args.max_tokens},
        hooks=hooks,
        n_parallel_tasks=args.n_parallel)

    trainer = UnifiedTrainer(
        backend, tasks,
        config=TrainerConfig(train_batch_size=args.batch_size,
                             rollout_n=args.rollout_n,
                             max_steps=args.steps,
                             episode_log_dir="episodes_swe"))
    trainer.fit()


if __name__ == "__main__":
    main()
