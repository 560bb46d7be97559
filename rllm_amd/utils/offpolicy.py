"""Off-policy diagnostics (K13; reference verl_backend.py:682-696):
cheap reductions comparing rollout-time logprobs with the recomputed
old-policy logprobs — KL estimate, perplexity ratio, Pearson correlation."""

from __future__ import annotations

import torch


@torch.no_grad()
def compute_offpolicy_metrics(old_logprob: torch.Tensor, rollout_logprob: torch.Tensor,
                              prefix: str = "offpolicy/") -> dict:
    """Both [N] fp32 over response tokens."""
    if old_logprob.numel() == 0:
        return {}
    d = old_logprob - rollout_logprob
    # k3 estimator of KL(rollout || old)
    kl = (torch.exp(d) - d - 1).mean()
    ppl_ratio = torch.exp(rollout_logprob.mean() - old_logprob.mean())
    x = rollout_logprob - rollout_logprob.mean()
    y = old_logprob - old_logprob.mean()
    denom = (x.norm() * y.norm()).clamp_min(1e-12)
    pearson = (x * y).sum() / denom
    return {
        f"{prefix}kl": float(kl),
        f"{prefix}ppl_ratio": float(ppl_ratio),
        f"{prefix}pearson": float(pearson),
        f"{prefix}abs_diff_mean": float(d.abs().mean()),
        f"{prefix}abs_diff_max": float(d.abs().max()),
    }
