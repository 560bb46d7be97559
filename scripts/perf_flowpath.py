#!/usr/bin/env python3
"""Gateway/flow-engine overhead at the flagship scale (VERDICT r1 weak #3:
bench.py calls engine.generate() directly; real training runs
AgentFlowEngine → HTTP gateway → trace enrich). This script runs the FULL
path — UnifiedTrainer + NativeBackend + gateway + @rollout flows — at the
bench shape (32 tasks × n=8 = 256 sessions, ~256-token prompts, 512 new
tokens, 1.5B) and reports per-step phase times to compare against bench.py.

  gpurun -- 'timeout 900 python scripts/perf_flowpath.py > gpurun_out/flowpath.log 2>&1'
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--tasks", type=int, default=32)
    ap.add_argument("--rollout-n", type=int, default=8)
    ap.add_argument("--prompt-bytes", type=int, default=230)
    ap.add_argument("--max-new-tokens", type=int, default=512)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--n-parallel", type=int, default=128)
    args = ap.parse_args()

    import httpx
    import torch

    import rllm_amd
    from rllm_amd.data.dataset import Dataset
    from rllm_amd.trainer.native_backend import NativeBackend
    from rllm_amd.trainer.policy import PolicyTrainerConfig
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer

    @rllm_amd.rollout
    def flow(task, config):
        # config.post uses the engine's SHARED pooled client — essential at
        # this concurrency (per-call clients throttle arrival to ~25/s)
        r = config.post("/chat/completions",
                        json={"model": config.model,
                              "messages": [{"role": "user", "content": str(task.instruction)}]})
        r.raise_for_status()
        return None

    @rllm_amd.evaluator
    def ev(task, episode):
        step = episode.trajectories[0].steps[-1]
        if not step.response_ids:
            return 0.0
        return float(sum(1 for t in step.response_ids if t % 2 == 0) / len(step.response_ids))

    # ByteTokenizer: 1 byte = 1 token, so prompt length ≈ prompt_bytes + template
    body = ("x" * args.prompt_bytes)
    tasks = Dataset([{"question": f"{i:03d}" + body, "id": str(i)}
                     for i in range(args.tasks)]).as_tasks(id_key="id")

    backend = NativeBackend(
        flow, ev, model_config=args.model,
        policy_config=PolicyTrainerConfig(lr=1e-6, kl_beta=1e-3, grad_clip=1.0,
                                          old_logprob_mode="alias",
                                          max_tokens_per_micro=32768),
        rollout_sampling_params={"temperature": 1.0, "top_p": 1.0,
                                 "max_tokens": args.max_new_tokens},
        n_parallel_tasks=args.n_parallel, seed=11)

    tcfg = TrainerConfig(total_epochs=100, train_batch_size=args.tasks,
                         rollout_n=args.rollout_n, max_steps=args.steps,
                         logger_backends=[])
    trainer = UnifiedTrainer(backend, tasks, config=tcfg)

    times = []
    orig = trainer._train_batch_async

    async def timed(batch):
        t0 = time.monotonic()
        m = await orig(batch)
        torch.cuda.synchronize()
        m["time/step_total_s"] = time.monotonic() - t0
        times.append({k: v for k, v in m.items() if k.startswith("time/") or k.startswith("batch/")})
        print(f"step: { {k: round(v, 2) for k, v in times[-1].items()} }", flush=True)
        st = backend.engine.stats
        if st["decode_calls"]:
            print(f"engine: decode_calls={st['decode_calls']} "
                  f"avg_decode_batch={st['decode_batch_sum']/st['decode_calls']:.1f} "
                  f"decode_tokens={st['decode_tokens']} "
                  f"prefill_calls={st['prefill_calls']} prefill_tokens={st['prefill_tokens']}",
                  flush=True)
            for k in st:
                st[k] = 0
        return m

    trainer._train_batch_async = timed
    trainer.fit()

    if len(times) > 1:
        steady = times[1:]
        gen = sum(t["time/gen_s"] for t in steady) / len(steady)
        upd = sum(t.get("time/update_s", 0.0) for t in steady) / len(steady)
        tot = sum(t["time/step_total_s"] for t in steady) / len(steady)
        toks = sum(t.get("batch/response_tokens", 0) for t in steady) / len(steady)
        print(f"\nsteady state: gen={gen:.2f}s update={upd:.2f}s total={tot:.2f}s "
              f"resp_tokens/step={toks:.0f} train_tok/s={toks / tot:,.0f}")
        print("compare: bench.py direct-engine path ≈ 2.25s rollout + 3.23s update "
              "(gateway/flow overhead = gen_s here minus bench rollout_s)")


if __name__ == "__main__":
    main()
