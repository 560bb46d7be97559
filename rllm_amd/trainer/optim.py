"""Flat-buffer AdamW driven by the fused HIP kernel (K11 in SURVEY.md §2.E).

All model params are re-materialized as views into ONE flat bf16 buffer;
their .grad fields are views into ONE flat bf16 grad buffer (autograd
accumulates in place). The optimizer keeps flat fp32 master/exp_avg/
exp_avg_sq. A step is then:
    RCCL all-reduce(flat grad)  [DP]
  → grad_sq_sum kernel          [global grad-norm, stays on device]
  → adamw_kernel                [moments + decoupled wd + bf16 write-back]
No per-tensor loops, no host sync.

LR schedules: constant / linear / cosine with warmup (reference
tinker_policy_trainer.py:416, base.yaml:82-84).
"""

from __future__ import annotations

import math

import torch

from rllm_amd import ops


def flatten_params(module: torch.nn.Module) -> tuple[torch.Tensor, torch.Tensor]:
    """Repoint every parameter of `module` into one flat bf16 buffer and
    pre-assign .grad views into a flat grad buffer. Returns (flat, flat_grad)."""
    params = [p for p in module.parameters() if p.requires_grad]
    total = sum(p.numel() for p in params)
    device = params[0].device
    flat = torch.empty(total, device=device, dtype=torch.bfloat16)
    flat_grad = torch.zeros(total, device=device, dtype=torch.bfloat16)
    off = 0
    for p in params:
        n = p.numel()
        flat[off : off + n].copy_(p.data.reshape(-1))
        p.data = flat[off : off + n].view(p.shape)
        p.grad = flat_grad[off : off + n].view(p.shape)
        off += n
    return flat, flat_grad


class FusedAdamW:
    def __init__(self, flat_param: torch.Tensor, flat_grad: torch.Tensor, *,
                 lr: float = 1e-6, betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0, grad_clip: float = 1.0,
                 lr_schedule: str = "constant", warmup_steps: int = 0,
                 total_steps: int = 0, min_lr_ratio: float = 0.0):
        self.flat_param = flat_param
        self.flat_grad = flat_grad
        self.master = flat_param.float()
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        self.base_lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.grad_clip = grad_clip
        self.lr_schedule = lr_schedule
        self.warmup_steps = warmup_steps
        self.total_steps = total_steps
        self.min_lr_ratio = min_lr_ratio
        self.step_count = 0

    def lr_at(self, step: int) -> float:
        lr = self.base_lr
        if self.warmup_steps > 0 and step < self.warmup_steps:
            return lr * (step + 1) / self.warmup_steps
        if self.lr_schedule == "constant" or self.total_steps <= 0:
            return lr
        progress = min(1.0, (step - self.warmup_steps) / max(1, self.total_steps - self.warmup_steps))
        if self.lr_schedule == "linear":
            return lr * (1 - (1 - self.min_lr_ratio) * progress)
        if self.lr_schedule == "cosine":
            return self.min_lr_ratio * lr + (1 - self.min_lr_ratio) * lr * 0.5 * (1 + math.cos(math.pi * progress))
        return lr

    def zero_grad(self):
        self.flat_grad.zero_()

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0) -> torch.Tensor:
        """Returns the (device) squared grad-norm tensor for metrics."""
        self.step_count += 1
        gnorm_sq = ops.grad_sq_sum(self.flat_grad, grad_scale) if self.grad_clip > 0 else None
        ops.adamw_step(
            self.flat_grad, self.master, self.exp_avg, self.exp_avg_sq, self.flat_param,
            gnorm_sq, lr=self.lr_at(self.step_count - 1), beta1=self.betas[0], beta2=self.betas[1],
            eps=self.eps, weight_decay=self.weight_decay, step=self.step_count,
            grad_clip=self.grad_clip, grad_scale=grad_scale)
        return gnorm_sq if gnorm_sq is not None else torch.zeros(1, device=self.flat_grad.device)

    def state_dict(self) -> dict:
        return {
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "step_count": self.step_count,
        }

    def load_state_dict(self, sd: dict):
        self.master.copy_(sd["master"])
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.step_count = int(sd["step_count"])
        self.flat_param.copy_(self.master.to(torch.bfloat16))
