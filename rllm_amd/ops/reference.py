"""Plain-PyTorch fp32 reference implementations of every HIP kernel.

Used by the GPU numerics tests (each kernel is compared against the fp32
reference on random inputs) and by nothing else — the GPU path never
falls back to these.
"""

from __future__ import annotations

import torch


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    xf = x.float()
    inv_rms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv_rms * w.float()).to(x.dtype)


def rope_ref(q: torch.Tensor, k: torch.Tensor, cos_tab: torch.Tensor, sin_tab: torch.Tensor,
             positions: torch.Tensor):
    """NeoX rotate-half. q [T,H,D]; cos/sin [P, D/2] fp32."""

    def _apply(x):
        xf = x.float()
        half = x.shape[-1] // 2
        c = cos_tab[positions.long()].unsqueeze(1)  # [T,1,half]
        s = sin_tab[positions.long()].unsqueeze(1)
        x1, x2 = xf[..., :half], xf[..., half:]
        out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
        return out.to(x.dtype)

    return _apply(q), _apply(k)


def swiglu_ref(gateup: torch.Tensor) -> torch.Tensor:
    g, u = gateup.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(g) * u).to(gateup.dtype)


def logprob_entropy_ref(hidden: torch.Tensor, lm_weight: torch.Tensor, targets: torch.Tensor,
                        temperature: float = 1.0):
    logits = (hidden.float() @ lm_weight.float().t()) / temperature
    logp = torch.log_softmax(logits, dim=-1)
    lp = logp.gather(1, targets.long().unsqueeze(1)).squeeze(1)
    p = logp.exp()
    entropy = -(p * logp).sum(-1)
    return lp, entropy


def grpo_loss_ref(logprob, old_logprob, ref_logprob, advantages, tis_w=None,
                  eps_lo=0.2, eps_hi=0.2, kl_beta=0.0):
    ratio = torch.exp(logprob - old_logprob)
    s1 = ratio * advantages
    s2 = torch.clamp(ratio, 1 - eps_lo, 1 + eps_hi) * advantages
    pg = -torch.min(s1, s2)
    clipped = (s1 > s2).float()
    loss = pg
    if kl_beta > 0 and ref_logprob is not None:
        d = ref_logprob - logprob
        loss = loss + kl_beta * (torch.exp(d) - d - 1)
    if tis_w is not None:
        loss = loss * tis_w
    return loss, clipped


def attention_ref(q, k, v, cu_seqlens, scale):
    """Causal varlen GQA attention, fp32. q [T,Hq,D], k/v [T,Hk,D]."""
    T, Hq, D = q.shape
    Hk = k.shape[1]
    G = Hq // Hk
    out = torch.zeros_like(q, dtype=torch.float32)
    for b in range(len(cu_seqlens) - 1):
        s0, s1 = int(cu_seqlens[b]), int(cu_seqlens[b + 1])
        n = s1 - s0
        qs = q[s0:s1].float()
        ks = k[s0:s1].float().repeat_interleave(G, dim=1)
        vs = v[s0:s1].float().repeat_interleave(G, dim=1)
        scores = torch.einsum("qhd,khd->hqk", qs, ks) * scale
        mask = torch.triu(torch.ones(n, n, device=q.device, dtype=torch.bool), diagonal=1)
        scores.masked_fill_(mask, -float("inf"))
        p = torch.softmax(scores, dim=-1)
        out[s0:s1] = torch.einsum("hqk,khd->qhd", p, vs)
    return out.to(q.dtype)


def adamw_ref(grad, master, m, v, *, lr, beta1, beta2, eps, weight_decay, step,
              grad_clip=0.0, grad_scale=1.0):
    g = grad.float() * grad_scale
    if grad_clip > 0:
        norm = g.norm()
        if norm > grad_clip:
            g = g * (grad_clip / (norm + 1e-6))
    m_new = beta1 * m + (1 - beta1) * g
    v_new = beta2 * v + (1 - beta2) * g * g
    mhat = m_new / (1 - beta1**step)
    vhat = v_new / (1 - beta2**step)
    master_new = master - lr * (mhat / (vhat.sqrt() + eps) + weight_decay * master)
    return master_new, m_new, v_new
