#!/usr/bin/env python3
"""Update-path microbenchmark: per-phase attribution for the GRPO update
(old-logprob fwd / ref fwd / policy fwd / backward / optimizer) plus an
effective-TFLOP/s estimate.

python scripts/perf_update.py [--seqs 256] [--seq-len 768] [--micro-tokens 32768]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np
import torch


def timed(fn):
    torch.cuda.synchronize()
    t0 = time.monotonic()
    out = fn()
    torch.cuda.synchronize()
    return out, time.monotonic() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seqs", type=int, default=256)
    ap.add_argument("--seq-len", type=int, default=768)
    ap.add_argument("--prompt-len", type=int, default=256)
    ap.add_argument("--micro-tokens", type=int, default=32768)
    ap.add_argument("--model", default="r1-distill-qwen-1.5b")
    ap.add_argument("--iters", type=int, default=2)
    ap.add_argument("--profile-bwd", action="store_true")
    args = ap.parse_args()

    from rllm_amd import ops
    from rllm_amd.models.config import get_model_config
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.trainer.batch import PackedRow, pack_rows, split_rows_token_balanced
    from rllm_amd.trainer.policy import PolicyTrainer, PolicyTrainerConfig

    cfg = get_model_config(args.model)
    model = QwenModel(cfg, device="cuda").init_random(seed=0)
    ref = QwenModel(cfg, device="cuda").init_random(seed=0)
    trainer = PolicyTrainer(model, ref, PolicyTrainerConfig(
        lr=1e-6, kl_beta=1e-3, max_tokens_per_micro=args.micro_tokens))

    rng = np.random.default_rng(0)
    S, P = args.seq_len, args.prompt_len
    rows = []
    for _ in range(args.seqs):
        toks = rng.integers(0, cfg.vocab_size, size=S).tolist()
        mask = [0] * P + [1] * (S - P)
        rows.append(PackedRow(toks, mask, [0.5] * S, [-1.0] * S))

    micros = split_rows_token_balanced(rows, args.micro_tokens)
    n_tokens = args.seqs * S
    n_loss = args.seqs * (S - P)

    def fwd_only(m, batch, rows_idx, grad=False):
        ctx = torch.enable_grad() if grad else torch.no_grad()
        with ctx:
            hidden = m.forward_train(batch.input_ids, batch.positions, batch.cu_seqlens)
            lp, _ = ops.chunked_logprob(hidden[rows_idx], m.lm_weight,
                                        batch.targets[rows_idx], want_entropy=False)
        return lp

    # warmup + phase timing on one micro
    batch = pack_rows(micros[0], device="cuda")
    rows_idx = batch.loss_mask.nonzero(as_tuple=True)[0]
    fwd_only(model, batch, rows_idx)
    torch.cuda.synchronize()

    _, t_fwd = timed(lambda: fwd_only(model, batch, rows_idx))
    _, t_ref = timed(lambda: fwd_only(ref, batch, rows_idx))

    def fwd_bwd():
        lp = fwd_only(model, batch, rows_idx, grad=True)
        lp.sum().backward()
    trainer.optim.zero_grad()
    _, t_fwdbwd = timed(fwd_bwd)

    if args.profile_bwd:
        from torch.profiler import ProfilerActivity, profile

        trainer.optim.zero_grad()
        with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
            fwd_bwd()
            torch.cuda.synchronize()
        print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=18,
                                        max_name_column_width=60))
        return
    trainer.optim.zero_grad()
    _, t_opt = timed(lambda: trainer.optim.step())

    micro_tokens = batch.input_ids.numel()
    body = 2 * 1.31e9 if "1.5b" in args.model else 2 * sum(
        p.numel() for n, p in model.named_parameters() if "embed" not in n and "lm_head" not in n)
    print(f"per-micro ({micro_tokens} tokens): fwd(old)={t_fwd*1e3:.0f}ms ref={t_ref*1e3:.0f}ms "
          f"fwd+bwd={t_fwdbwd*1e3:.0f}ms optim={t_opt*1e3:.0f}ms")
    flops_fwd = 2 * 1.78e9 * micro_tokens
    print(f"  fwd eff ~{flops_fwd/t_fwd/1e12:.0f} TF/s; fwd+bwd eff ~{3*flops_fwd/t_fwdbwd/1e12:.0f} TF/s")

    # full update_policy
    def full():
        def old_lp_fn(b, ri):
            return fwd_only(model, b, ri)
        return trainer.update_policy(rows, old_logprob_fn=old_lp_fn)
    for _ in range(args.iters):
        _, t_full = timed(full)
    print(f"full update: {t_full*1e3:.0f}ms for {n_tokens} tokens ({n_loss} loss tokens) "
          f"-> {n_tokens/t_full:,.0f} tok/s ; micros={len(micros)}")


if __name__ == "__main__":
    main()
