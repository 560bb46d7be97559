"""PolicyTrainer — the native policy-update path (replaces verl FSDP
workers).

One GPU holds: actor (bf16, flat), frozen reference policy (bf16), fp32
optimizer state, and — colocated — the rollout engine SHARING the actor's
weight tensors (288 GB HBM). DP across GPUs = RCCL all-reduce of the flat
gradient buffer.

update_policy semantics mirror the reference defaults
(_generated_agent_ppo_trainer.yaml): ratio-clip 0.2 (asymmetric high clip
supported), KL-in-loss low_var_kl, loss_agg token-mean with the GLOBAL
response-token denominator, dynamic token-balanced micro-batches with
all_reduce(MAX) of the micro-batch count across DP (patch.py:23-66
deadlock guard).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass

import torch

from rllm_amd import ops
from rllm_amd.parallel import dist as pdist
from rllm_amd.trainer.algorithms.config import AlgorithmConfig
from rllm_amd.trainer.batch import PackedRow, TrainBatch, pack_rows, split_rows_token_balanced
from rllm_amd.trainer.optim import FusedAdamW, flatten_params

logger = logging.getLogger(__name__)


@dataclass
class PolicyTrainerConfig:
    lr: float = 1e-6
    betas: tuple = (0.9, 0.999)
    weight_decay: float = 0.0
    grad_clip: float = 1.0
    eps_clip: float = 0.2
    eps_clip_high: float | None = None
    kl_beta: float = 0.0
    max_tokens_per_micro: int = 16384
    loss_agg_mode: str = "token-mean"
    tis_mode: str | None = None     # None | "token" | "sequence"
    tis_cap: float = 5.0
    bypass_mode: bool = False        # π_old := π_rollout (skip recompute)
    # old-logprob source: "recompute" = separate no-grad forward (reference
    # behavior, verl_backend.py:639); "alias" = lp.detach() — bit-identical
    # to recompute when there is exactly ONE optimizer step per mini-batch
    # (deterministic kernels, same weights) but saves a full forward pass;
    # "rollout" = behaves like bypass_mode.
    old_logprob_mode: str = "recompute"
    # rollout sampling temperature: the sampler records logprobs of
    # logits/T (sampling.hip), so every training-side logprob (new, old,
    # ref) must be computed at the SAME temperature or the IS ratios and
    # KL compare mismatched distributions (reference propagates it via
    # meta_info['temperature'], verl_backend.py:612).
    temperature: float = 1.0
    entropy_chunk: int = 16384
    # run the frozen-ref no-grad forward of micro i+1 on a second HIP
    # stream, overlapped with micro i's backward (single-GPU latency hide)
    overlap_ref_stream: bool = True
    # Ulysses sequence-parallel update (SURVEY §2.D SP; parallel/ulysses.py):
    # ALL ranks hold the same rows; each runs its token shard of every
    # micro with all-to-alls around attention. Long-context updates only.
    sequence_parallel: bool = False
    use_ref: bool = True
    # LoRA training: the KL reference is the actor's own base weights with
    # adapters disabled (models/lora.py) — no second model copy.
    ref_from_lora_base: bool = False

    @classmethod
    def from_algorithm_config(cls, a: AlgorithmConfig, **over) -> "PolicyTrainerConfig":
        return cls(
            eps_clip=a.eps_clip,
            eps_clip_high=a.eps_clip_high,
            kl_beta=a.kl_beta,
            loss_agg_mode=a.loss_agg_mode or "token-mean",
            tis_mode=a.rollout_correction.tis_mode,
            tis_cap=a.rollout_correction.tis_cap,
            bypass_mode=bool(a.rollout_correction.bypass_mode),
            use_ref=a.kl_beta > 0,
            **over,
        )


def tis_weights(old_lp: torch.Tensor, rollout_lp: torch.Tensor, *,
                mode: str | None, cap: float,
                seq_ids: torch.Tensor | None = None) -> torch.Tensor | None:
    """Truncated importance-sampling correction weights
    (reference verl_backend.py:663-676).

    mode "token": per-token exp(old - rollout), clamped to cap.
    mode "sequence": one weight per sequence, exp(sum of per-token
    log-ratios over that sequence), clamped, broadcast back to its tokens
    (requires seq_ids: for each loss row, which packed sequence it is in).
    """
    if mode is None:
        return None
    log_ratio = old_lp - rollout_lp
    if mode == "sequence":
        if seq_ids is None:
            raise ValueError("sequence-level TIS needs seq_ids")
        n_seq = int(seq_ids.max().item()) + 1 if seq_ids.numel() else 0
        seq_sum = torch.zeros(n_seq, dtype=log_ratio.dtype, device=log_ratio.device)
        seq_sum.index_add_(0, seq_ids, log_ratio)
        w_seq = torch.exp(seq_sum).clamp(max=cap)
        return w_seq[seq_ids]
    if mode != "token":
        raise ValueError(f"unknown tis_mode {mode!r}")
    return torch.exp(log_ratio).clamp(max=cap)


class PolicyTrainer:
    def __init__(self, model, ref_model=None, config: PolicyTrainerConfig | None = None):
        self.model = model
        self.ref_model = ref_model
        self.cfg = config or PolicyTrainerConfig()
        self.device = next(model.parameters()).device
        self.flat_param, self.flat_grad = flatten_params(model)
        self.optim = FusedAdamW(
            self.flat_param, self.flat_grad,
            lr=self.cfg.lr, betas=self.cfg.betas, weight_decay=self.cfg.weight_decay,
            grad_clip=self.cfg.grad_clip)
        self.weight_version = 0

    # ------------------------------------------------------------------
    @torch.no_grad()
    def compute_logprobs(self, batch: TrainBatch, model=None, want_entropy: bool = False):
        """Forward pass → logprob of each loss token (fp32, [n_resp]);
        optionally mean entropy. Used for old-logprob and ref-logprob
        (reference compute_log_prob / compute_ref_log_prob RPCs)."""
        model = model or self.model
        hidden = model.forward_train(batch.input_ids, batch.positions, batch.cu_seqlens)
        rows = batch.loss_mask.nonzero(as_tuple=True)[0]
        h = hidden[rows]
        tgt = batch.targets[rows]
        lp, ent = ops.chunked_logprob(h, model.lm_weight, tgt,
                                      chunk=self.cfg.entropy_chunk,
                                      temperature=self.cfg.temperature,
                                      want_entropy=want_entropy)
        return lp, (ent.mean().item() if ent is not None else None)

    # ------------------------------------------------------------------
    def _tis_weights(self, old_lp: torch.Tensor, rollout_lp: torch.Tensor,
                     seq_ids: torch.Tensor | None = None) -> torch.Tensor | None:
        return tis_weights(old_lp, rollout_lp, mode=self.cfg.tis_mode,
                           cap=self.cfg.tis_cap, seq_ids=seq_ids)

    def update_policy(self, rows: list[PackedRow], old_logprob_fn=None) -> dict:
        """One optimizer step over a mini-batch of packed rows.

        Pipeline per micro-batch: forward (custom kernels) → chunked
        logprob on loss rows → fused GRPO loss (loss + dlp in one kernel)
        → backward into the flat grad buffer. Then RCCL all-reduce +
        fused AdamW. Returns metrics.
        """
        cfg = self.cfg
        if cfg.sequence_parallel and pdist.get_world_size() > 1:
            return self._update_policy_sp(rows)
        micros = split_rows_token_balanced(rows, cfg.max_tokens_per_micro)
        # DP desync guard: every rank must run the same number of micro
        # fwd/bwd? No — grads accumulate locally; only the all-reduce at the
        # end must match. But the GLOBAL token denominator must agree:
        n_local_tokens = sum(sum(r.response_mask) for r in rows)
        n_global_tokens = pdist.all_reduce_scalar(float(n_local_tokens), op="sum")
        world = pdist.get_world_size()
        # seq-mean denominator: ONE collective, BEFORE the micro loop — ranks
        # have different micro counts, so any collective inside it deadlocks
        n_global_rows = (pdist.all_reduce_scalar(float(len(rows)), op="sum")
                         if cfg.loss_agg_mode == "seq-mean-token-mean" else 0.0)
        if n_global_tokens <= 0:
            logger.warning("update_policy called with zero response tokens; skipping step")
            return {"actor/skipped": 1.0}

        self.optim.zero_grad()
        offpolicy_metrics: dict = {}
        tot_loss = 0.0
        tot_pg = 0.0
        tot_kl_count = 0
        tot_clip = 0.0
        ent_sum, ent_n = 0.0, 0

        # pack every micro up front so the ref-prefetch stream can read
        # batch i+1's tensors while micro i is still in backward
        packed = [pack_rows(m, device=str(self.device)) for m in micros]
        use_ref_any = cfg.use_ref and (cfg.ref_from_lora_base or self.ref_model is not None)
        # overlap the (no-grad) frozen-reference forward of micro i+1 with
        # micro i's backward on a second HIP stream: the ref path is pure
        # torch.matmul + stateless HIP kernels (no shared workspace), so the
        # streams don't contend on state — only on CUs
        overlap = (cfg.overlap_ref_stream and use_ref_any and len(packed) > 1
                   and getattr(self.device, "type", str(self.device)) == "cuda")
        ref_stream = torch.cuda.Stream(device=self.device) if overlap else None
        if overlap:
            ref_stream.wait_stream(torch.cuda.current_stream())  # packs visible
        prefetched: dict[int, torch.Tensor] = {}

        @torch.no_grad()
        def ref_lp_of(i: int) -> torch.Tensor:
            b = packed[i]
            ridx = b.loss_mask.nonzero(as_tuple=True)[0]
            if cfg.ref_from_lora_base:
                from rllm_amd.models import lora as _lora

                with _lora.disabled(self.model):
                    rh = self.model.forward_train(b.input_ids, b.positions, b.cu_seqlens)
                    rlp, _ = ops.chunked_logprob(rh[ridx], self.model.lm_weight,
                                                 b.targets[ridx], chunk=cfg.entropy_chunk,
                                                 temperature=cfg.temperature, want_entropy=False)
            else:
                rh = self.ref_model.forward_train(b.input_ids, b.positions, b.cu_seqlens)
                rlp, _ = ops.chunked_logprob(rh[ridx], self.ref_model.lm_weight,
                                             b.targets[ridx], chunk=cfg.entropy_chunk,
                                             temperature=cfg.temperature, want_entropy=False)
            return rlp

        for mi, batch in enumerate(packed):
            micro = micros[mi]
            hidden = self.model.forward_train(batch.input_ids, batch.positions, batch.cu_seqlens)
            rows_idx = batch.loss_mask.nonzero(as_tuple=True)[0]
            h = hidden[rows_idx]
            tgt = batch.targets[rows_idx]
            adv = batch.advantages[rows_idx]
            rollout_lp = batch.rollout_logprobs[rows_idx]

            lp, ent = ops.chunked_logprob(h, self.model.lm_weight, tgt,
                                          chunk=cfg.entropy_chunk,
                                          temperature=cfg.temperature, want_entropy=True)
            if ent is not None:
                ent_sum += float(ent.sum())
                ent_n += ent.numel()

            with torch.no_grad():
                if cfg.bypass_mode or cfg.old_logprob_mode == "rollout":
                    old_lp = rollout_lp
                elif cfg.old_logprob_mode == "alias":
                    old_lp = lp.detach()
                elif old_logprob_fn is None:
                    # recompute mode REQUIRES the callback — silently falling
                    # back to rollout logprobs would give bypass semantics
                    # without the caller asking for them
                    raise ValueError(
                        "old_logprob_mode='recompute' needs old_logprob_fn; pass one "
                        "or set old_logprob_mode='rollout'/bypass_mode=True explicitly")
                else:
                    old_lp = old_logprob_fn(batch, rows_idx)
                ref_lp = None
                if use_ref_any:
                    if mi in prefetched:
                        torch.cuda.current_stream().wait_stream(ref_stream)
                        ref_lp = prefetched.pop(mi)
                        ref_lp.record_stream(torch.cuda.current_stream())
                    else:
                        ref_lp = ref_lp_of(mi)
                seq_ids = None
                if cfg.tis_mode == "sequence" or cfg.loss_agg_mode == "seq-mean-token-mean":
                    # right=True: a loss row landing exactly on a sequence
                    # boundary (single-token prompt) belongs to the sequence
                    # that STARTS there, not the previous one
                    seq_ids = torch.bucketize(
                        rows_idx, torch.tensor(batch.cu_seqlens[1:-1], device=rows_idx.device),
                        right=True)
                tis_w = self._tis_weights(old_lp, rollout_lp, seq_ids=seq_ids)

            if mi == 0:
                from rllm_amd.utils.offpolicy import compute_offpolicy_metrics

                offpolicy_metrics = compute_offpolicy_metrics(old_lp, rollout_lp)
            eps_hi = cfg.eps_clip_high if cfg.eps_clip_high is not None else cfg.eps_clip
            loss_tok, clipped = ops.grpo_loss_per_token(
                lp, old_lp, ref_lp, adv, tis_w,
                eps_lo=cfg.eps_clip, eps_hi=eps_hi, kl_beta=cfg.kl_beta)

            if cfg.loss_agg_mode == "token-mean":
                # global-token-mean: SUM-all-reduce of grads then NO rescale
                loss = loss_tok.sum() / n_global_tokens
            elif cfg.loss_agg_mode == "seq-mean-token-mean":
                # per-sequence token mean, then mean over the GLOBAL sequence
                # count (computed once above) — one index_add per micro, not
                # one kernel launch per sequence
                n_seq = int(seq_ids.max().item()) + 1
                zeros = torch.zeros(n_seq, dtype=loss_tok.dtype, device=loss_tok.device)
                seq_sum = zeros.index_add(0, seq_ids, loss_tok)
                seq_cnt = zeros.index_add(0, seq_ids, torch.ones_like(loss_tok.detach()))
                loss = (seq_sum / seq_cnt.clamp(min=1.0)).sum() / max(1.0, n_global_rows)
            else:
                loss = loss_tok.sum() / n_global_tokens
            loss.backward()
            # prefetch the NEXT micro's frozen-ref forward concurrently with
            # this micro's backward
            if overlap and mi + 1 < len(packed) and (mi + 1) not in prefetched:
                with torch.cuda.stream(ref_stream):
                    prefetched[mi + 1] = ref_lp_of(mi + 1)
            tot_loss += float(loss.detach())
            tot_pg += float(loss_tok.detach().sum())
            tot_clip += float(clipped.sum())
            tot_kl_count += int(loss_tok.numel())

        # C1: DP gradient all-reduce (SUM; the global denominator already
        # normalizes, so no 1/world rescale for token-mean)
        if world > 1:
            pdist.all_reduce_sum_(self.flat_grad)
        gnorm_sq = self.optim.step(grad_scale=1.0)
        self.weight_version += 1

        metrics = {
            "actor/loss": tot_loss,
            "actor/pg_loss_tok_mean": tot_pg / max(1, n_local_tokens),
            "actor/clipfrac": tot_clip / max(1, tot_kl_count),
            "actor/entropy": ent_sum / max(1, ent_n),
            "actor/grad_norm": float(gnorm_sq.sqrt().item()),
            "actor/lr": self.optim.lr_at(self.optim.step_count - 1),
            "actor/n_micro_batches": len(micros),
            "actor/n_response_tokens": n_local_tokens,
        }
        metrics.update(offpolicy_metrics)
        return metrics

    # ------------------------------------------------------------------
    def _update_policy_sp(self, rows: list[PackedRow]) -> dict:
        """Ulysses SP update (trainer-side plumbing, TODO r1 #10): every
        rank holds the SAME rows; each packs the same micros (padded to the
        SP degree) and runs forward/backward on ITS token shard with
        all-to-alls around attention (models/qwen.py sp_group). Gradients
        are partial sums over disjoint loss-token shards → the same
        all-reduce(SUM) as DP completes them.

        old-logprob source must be 'alias' or 'rollout' here (a recompute
        callback would need its own sharded forward)."""
        import torch.distributed as tdist

        from rllm_amd.parallel import ulysses

        cfg = self.cfg
        if cfg.old_logprob_mode == "recompute" and not cfg.bypass_mode:
            raise ValueError("sequence_parallel update needs old_logprob_mode "
                             "'alias' or 'rollout' (no recompute callback)")
        if cfg.loss_agg_mode != "token-mean":
            raise ValueError("sequence_parallel update supports token-mean only")
        if cfg.use_ref and cfg.ref_from_lora_base:
            raise ValueError("sequence_parallel + LoRA-base KL is not wired; "
                             "pass an explicit ref_model or disable use_ref")
        P = pdist.get_world_size()
        rank = pdist.get_rank()
        group = self.model.sp_group or tdist.group.WORLD
        self.model.sp_group = group
        if self.ref_model is not None:
            self.ref_model.sp_group = group

        micros = split_rows_token_balanced(rows, cfg.max_tokens_per_micro)
        # C3 (reference patch.py:23-66): attention all-to-alls run INSIDE
        # each micro — ranks MUST agree on the micro count or the group
        # deadlocks. Rows are replicated so the schedules match; verify
        # with an all_reduce(MAX) before entering the loop.
        n_micro_max = pdist.all_reduce_scalar(float(len(micros)), op="max")
        if n_micro_max != float(len(micros)):
            raise RuntimeError(
                f"SP micro-batch schedule diverged: local {len(micros)} vs max {n_micro_max} "
                "(rows must be identical on every SP rank)")

        # global denominator: rows are the FULL batch on every rank
        n_total_tokens = float(sum(sum(r.response_mask) for r in rows))
        if n_total_tokens <= 0:
            return {"actor/skipped": 1.0}

        self.optim.zero_grad()
        tot_loss = 0.0
        tot_clip = 0.0
        tot_kl_count = 0
        ent_sum, ent_n = 0.0, 0
        n_local_tokens = 0

        eps_hi = cfg.eps_clip_high if cfg.eps_clip_high is not None else cfg.eps_clip
        for micro in micros:
            batch = pack_rows(micro, device=str(self.device), pad_to_multiple=P)
            T = batch.input_ids.shape[0]
            sl = ulysses.shard_slice(T, rank, P)
            hidden = self.model.forward_train(batch.input_ids[sl], batch.positions[sl],
                                              batch.cu_seqlens)
            lm = batch.loss_mask[sl]
            rows_idx = lm.nonzero(as_tuple=True)[0]
            if rows_idx.numel() == 0:
                # this shard holds no loss tokens — backward still must run
                # (the all-to-alls have gradient paths on every rank)
                loss = (hidden.float() * 0).sum()
                loss.backward()
                continue
            tgt = batch.targets[sl][rows_idx]
            adv = batch.advantages[sl][rows_idx]
            rollout_lp = batch.rollout_logprobs[sl][rows_idx]
            lp, ent = ops.chunked_logprob(hidden[rows_idx], self.model.lm_weight, tgt,
                                          chunk=cfg.entropy_chunk,
                                          temperature=cfg.temperature, want_entropy=True)
            if ent is not None:
                ent_sum += float(ent.sum())
                ent_n += ent.numel()
            with torch.no_grad():
                old_lp = rollout_lp if (cfg.bypass_mode or cfg.old_logprob_mode == "rollout") \
                    else lp.detach()
                ref_lp = None
                if cfg.use_ref and self.ref_model is not None:
                    rh = self.ref_model.forward_train(batch.input_ids[sl], batch.positions[sl],
                                                      batch.cu_seqlens)
                    ref_lp, _ = ops.chunked_logprob(rh[rows_idx], self.ref_model.lm_weight,
                                                    tgt, chunk=cfg.entropy_chunk,
                                                    temperature=cfg.temperature,
                                                    want_entropy=False)
                tis_w = self._tis_weights(old_lp, rollout_lp)
            loss_tok, clipped = ops.grpo_loss_per_token(
                lp, old_lp, ref_lp, adv, tis_w,
                eps_lo=cfg.eps_clip, eps_hi=eps_hi, kl_beta=cfg.kl_beta)
            loss = loss_tok.sum() / n_total_tokens
            loss.backward()
            tot_loss += float(loss.detach())
            tot_clip += float(clipped.sum())
            tot_kl_count += int(loss_tok.numel())
            n_local_tokens += int(rows_idx.numel())

        # shard grads are partial sums over disjoint loss tokens
        pdist.all_reduce_sum_(self.flat_grad)
        gnorm_sq = self.optim.step(grad_scale=1.0)
        self.weight_version += 1
        loss_global = pdist.all_reduce_scalar(tot_loss, op="sum")
        return {
            "actor/loss": loss_global,
            "actor/clipfrac": tot_clip / max(1, tot_kl_count),
            "actor/entropy": ent_sum / max(1, ent_n),
            "actor/grad_norm": float(gnorm_sq.sqrt().item()),
            "actor/lr": self.optim.lr_at(self.optim.step_count - 1),
            "actor/n_micro_batches": len(micros),
            "actor/n_response_tokens": n_local_tokens,
            "actor/sp_degree": P,
        }

    # ------------------------------------------------------------------
    def state_dict(self) -> dict:
        return {
            "model": {k: v for k, v in self.model.state_dict().items()},
            "optim": self.optim.state_dict(),
            "weight_version": self.weight_version,
        }

    def load_state_dict(self, sd: dict):
        self.model.load_state_dict(sd["model"])
        # only trainable params live in the flat buffer (all of them for a
        # full finetune; adapters only under LoRA)
        self.flat_param.copy_(torch.cat(
            [p.data.reshape(-1) for p in self.model.parameters() if p.requires_grad]))
        self.optim.load_state_dict(sd["optim"])
        self.weight_version = int(sd.get("weight_version", 0))
