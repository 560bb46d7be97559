"""Qwen2-family decoder, MI355X-native.

Three forward paths over one set of bf16 weights:

* ``forward_train`` — packed varlen, autograd-enabled. Elementwise/norm ops
  are the in-repo HIP kernels (rmsnorm / rope / swiglu autograd wrappers);
  projections are hipBLASLt GEMMs (torch.matmul); attention is a chunked
  bf16-matmul + fp32-softmax composition (the update-path attention; the
  hand-written flash kernels serve the rollout path — SURVEY.md §2.E K5/K7
  vs K1/K2). The LM head goes through ops.chunked_logprob so the [T, vocab]
  logits matrix is never materialized.
* ``forward_prefill`` — no-grad rollout prefill: flash_prefill HIP kernel +
  reshape_and_cache into KV pages.
* ``forward_decode`` — no-grad one-token step over the paged KV cache.

Weight layout is fused (qkv_proj, gate_up_proj) for GEMM efficiency;
``load_hf_state_dict`` maps HuggingFace Qwen2 checkpoints onto it.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from rllm_amd import ops
from rllm_amd.models.config import ModelConfig


def _linear(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor | None = None) -> torch.Tensor:
    return torch.nn.functional.linear(x, w, b)


def train_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    cu_seqlens: list[int], scale: float, q_chunk: int = 1024) -> torch.Tensor:
    """Causal GQA attention for the update path: bf16 GEMMs (hipBLASLt) with
    fp32 softmax, chunked over query blocks to bound the score matrix.

    q [T, Hq, D], k/v [T, Hk, D] packed varlen; returns [T, Hq, D] bf16.
    Uniform-length batches (the common RL case: equal-length synthetic or
    length-bucketed micro-batches) take a fully batched path — one set of
    GEMMs per layer instead of a per-sequence Python loop.
    """
    T, Hq, D = q.shape
    Hk = k.shape[1]
    G = Hq // Hk

    lens = [cu_seqlens[i + 1] - cu_seqlens[i] for i in range(len(cu_seqlens) - 1)]
    if len(set(lens)) == 1 and len(lens) > 1:
        return _train_attention_uniform(q, k, v, len(lens), lens[0], scale, q_chunk)

    out = torch.empty_like(q)
    for b in range(len(cu_seqlens) - 1):
        s0, s1 = cu_seqlens[b], cu_seqlens[b + 1]
        n = s1 - s0
        if n == 0:
            continue
        qs = q[s0:s1].permute(1, 0, 2)                       # [Hq, n, D]
        ks = k[s0:s1].permute(1, 0, 2)                       # [Hk, n, D]
        vs = v[s0:s1].permute(1, 0, 2)
        if G > 1:
            ks = ks.repeat_interleave(G, dim=0)
            vs = vs.repeat_interleave(G, dim=0)
        rows = []
        for c0 in range(0, n, q_chunk):
            c1 = min(n, c0 + q_chunk)
            scores = torch.matmul(qs[:, c0:c1], ks.transpose(1, 2)).float() * scale  # [Hq, c, n]
            mask = torch.arange(n, device=q.device).unsqueeze(0) > torch.arange(c0, c1, device=q.device).unsqueeze(1)
            scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
            p = torch.softmax(scores, dim=-1).to(q.dtype)
            rows.append(torch.matmul(p, vs))                 # [Hq, c, D]
        out[s0:s1] = torch.cat(rows, dim=1).permute(1, 0, 2)
    return out


def _train_attention_uniform(q, k, v, B: int, S: int, scale: float, q_chunk: int) -> torch.Tensor:
    """Batched path for B equal-length sequences: [B*Hq, S, D] bmm with one
    causal mask shared across the batch."""
    Hq, D = q.shape[1], q.shape[2]
    Hk = k.shape[1]
    G = Hq // Hk
    qs = q.view(B, S, Hq, D).permute(0, 2, 1, 3)          # [B, Hq, S, D]
    ks = k.view(B, S, Hk, D).permute(0, 2, 1, 3)
    vs = v.view(B, S, Hk, D).permute(0, 2, 1, 3)
    if G > 1:
        ks = ks.repeat_interleave(G, dim=1)
        vs = vs.repeat_interleave(G, dim=1)
    qs = qs.reshape(B * Hq, S, D)
    ks = ks.reshape(B * Hq, S, D)
    vs = vs.reshape(B * Hq, S, D)
    outs = []
    pos = torch.arange(S, device=q.device)
    for c0 in range(0, S, q_chunk):
        c1 = min(S, c0 + q_chunk)
        scores = torch.matmul(qs[:, c0:c1], ks.transpose(1, 2)).float() * scale  # [BH, c, S]
        mask = pos.unsqueeze(0) > pos[c0:c1].unsqueeze(1)
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        p = torch.softmax(scores, dim=-1).to(q.dtype)
        outs.append(torch.matmul(p, vs))                   # [BH, c, D]
    o = torch.cat(outs, dim=1).view(B, Hq, S, D).permute(0, 2, 1, 3)  # [B, S, Hq, D]
    return o.reshape(B * S, Hq, D).contiguous()


class QwenLayer(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        H = cfg.hidden_size
        self.cfg = cfg
        qkv_out = cfg.q_size + 2 * cfg.kv_size
        self.input_layernorm = nn.Parameter(torch.ones(H, dtype=torch.bfloat16))
        self.qkv_proj = nn.Parameter(torch.empty(qkv_out, H, dtype=torch.bfloat16))
        self.qkv_bias = nn.Parameter(torch.zeros(qkv_out, dtype=torch.bfloat16)) if cfg.qkv_bias else None
        self.o_proj = nn.Parameter(torch.empty(H, cfg.q_size, dtype=torch.bfloat16))
        self.post_attention_layernorm = nn.Parameter(torch.ones(H, dtype=torch.bfloat16))
        self.gate_up_proj = nn.Parameter(torch.empty(2 * cfg.intermediate_size, H, dtype=torch.bfloat16))
        self.down_proj = nn.Parameter(torch.empty(H, cfg.intermediate_size, dtype=torch.bfloat16))

    # LoRA hooks (models/lora.py): `lora` (a ParameterDict of A/B pairs) is
    # registered by inject_lora; no class default — a class attribute would
    # shadow nn.Module submodule lookup. `_lora_on` is stamped per-forward
    # by QwenModel from its enabled flag.
    lora_scale: float = 1.0
    _lora_on: bool = False
    # Rollout tensor parallelism (parallel/tp.py enable_tp): o_proj and
    # down_proj are row-parallel — their partial outputs are all-reduced.
    tp_group = None

    def _tp_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.tp_group is not None:
            import torch.distributed as dist

            dist.all_reduce(t, group=self.tp_group)
        return t

    def _lin(self, x: torch.Tensor, name: str, b: torch.Tensor | None = None) -> torch.Tensor:
        y = _linear(x, getattr(self, name), b)
        if self._lora_on and f"{name}_A" in self.lora:
            from rllm_amd.models import lora as _lora

            y = y + _lora.lora_delta(self, name, x)
        return y

    def _qkv(self, hidden: torch.Tensor, positions: torch.Tensor, cos_t, sin_t):
        cfg = self.cfg
        T = hidden.shape[0]
        x = ops.rmsnorm(hidden, self.input_layernorm, cfg.rms_eps)
        qkv = self._lin(x, "qkv_proj", self.qkv_bias)
        q, k, v = qkv.split([cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1)
        q = q.view(T, cfg.num_heads, cfg.head_dim)
        k = k.view(T, cfg.num_kv_heads, cfg.head_dim)
        v = v.view(T, cfg.num_kv_heads, cfg.head_dim).contiguous()
        q, k = ops.rope(q, k, cos_t, sin_t, positions)
        return q, k, v

    def _finish(self, hidden: torch.Tensor, attn_out: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        T = hidden.shape[0]
        hidden = hidden + self._tp_reduce(self._lin(attn_out.reshape(T, cfg.q_size), "o_proj"))
        x = ops.rmsnorm(hidden, self.post_attention_layernorm, cfg.rms_eps)
        mlp = self._tp_reduce(self._lin(ops.swiglu(self._lin(x, "gate_up_proj")), "down_proj"))
        return hidden + mlp

    # -- training path -------------------------------------------------------
    def forward_train(self, hidden, positions, cu_seqlens, cos_t, sin_t,
                      flash_tiles=None, sp_group=None):
        """With sp_group (Ulysses, parallel/ulysses.py): hidden/positions are
        this rank's token shard, cu_seqlens/flash_tiles describe the FULL
        packed batch; two all-to-alls around attention give full-sequence
        visibility per head shard."""
        q, k, v = self._qkv(hidden, positions, cos_t, sin_t)
        scale = 1.0 / math.sqrt(self.cfg.head_dim)

        def attn_fn(qh, kh, vh):
            if flash_tiles is not None:
                return ops.flash_attention_train(qh, kh, vh, *flash_tiles, scale)
            return train_attention(qh, kh, vh, cu_seqlens, scale)

        if sp_group is not None:
            from rllm_amd.parallel import ulysses

            attn = ulysses.ulysses_attention(q, k, v, attn_fn, sp_group)
        else:
            attn = attn_fn(q, k, v)
        return self._finish(hidden, attn)

    # -- rollout paths (no grad) ----------------------------------------------
    @torch.no_grad()
    def forward_prefill(self, hidden, positions, cos_t, sin_t, tiles, kv_cache, slot_mapping, layer_idx):
        q, k, v = self._qkv(hidden, positions, cos_t, sin_t)
        k_pages, v_pages = kv_cache[layer_idx]
        ops.reshape_and_cache(k, v, k_pages, v_pages, slot_mapping)
        attn = ops.flash_prefill(q, k, v, tiles[0], tiles[1], tiles[2],
                                 1.0 / math.sqrt(self.cfg.head_dim))
        return self._finish(hidden, attn)

    @torch.no_grad()
    def forward_prefill_cached(self, hidden, positions, cos_t, sin_t, tiles, kv_cache,
                               slot_suffix, slot_full, suffix_rows, T_full, layer_idx):
        """Prefix-cached prefill: `hidden`/`positions` cover only the SUFFIX
        tokens (absolute positions); cached prefix K/V already live in the
        pages. Suffix K/V are written, full-length K/V are gathered back
        from pages, and the flash kernel runs with q-tiles restricted to the
        suffix (tile_row0 starts past the cached region)."""
        cfg = self.cfg
        q, k, v = self._qkv(hidden, positions, cos_t, sin_t)
        k_pages, v_pages = kv_cache[layer_idx]
        ops.reshape_and_cache(k, v, k_pages, v_pages, slot_suffix)
        k_full, v_full = ops.gather_cache(k_pages, v_pages, slot_full)
        q_full = q.new_zeros(T_full, cfg.num_heads, cfg.head_dim)
        q_full.index_copy_(0, suffix_rows, q)
        attn_full = ops.flash_prefill(q_full, k_full, v_full, tiles[0], tiles[1], tiles[2],
                                      1.0 / math.sqrt(cfg.head_dim))
        attn = attn_full.index_select(0, suffix_rows)
        return self._finish(hidden, attn)

    @torch.no_grad()
    def forward_decode_fused(self, h, delta, positions, cos_t, sin_t, kv_cache,
                             slot_mapping, block_tables, seq_lens, layer_idx):
        """Decode layer with fused residual chaining: returns the next delta.
        Kernel sequence: add_rmsnorm -> qkv GEMM -> qkv_rope_cache ->
        paged_decode -> o GEMM -> add_rmsnorm -> gate_up GEMM -> swiglu ->
        down GEMM (9 launches/layer)."""
        cfg = self.cfg
        T = h.shape[0]
        k_pages, v_pages = kv_cache[layer_idx]
        fp8 = getattr(self, "_fp8", None)
        # fused-quant fp8 path (TODO r1 #12): the producing kernels emit
        # e4m3 directly for the qkv/gate_up/down GEMM inputs — no standalone
        # quant launches. o_proj keeps the delayed-quant inside _lin_decode
        # (its input comes from paged_decode's merge).
        fuse8 = (fp8 is not None and not self._lora_on
                 and all(n in fp8 for n in ("qkv_proj", "gate_up_proj", "down_proj")))
        if fuse8:
            w8q, swq, sq, aq = fp8["qkv_proj"]
            x8 = ops.add_rmsnorm_fp8_(h, delta, self.input_layernorm, cfg.rms_eps, sq, aq)
            qkv = ops.fp8_mm_prequant(x8, w8q, sq, swq)
        else:
            x = ops.add_rmsnorm_(h, delta, self.input_layernorm, cfg.rms_eps)
            qkv = self._lin_decode(x, "qkv_proj")
        q = ops.qkv_rope_cache(qkv, self.qkv_bias, k_pages, v_pages, cos_t, sin_t,
                               positions, slot_mapping, cfg.num_heads, cfg.num_kv_heads)
        attn = ops.paged_decode(q, k_pages, v_pages, block_tables, seq_lens,
                                1.0 / math.sqrt(cfg.head_dim))
        attn_delta = self._tp_reduce(self._lin_decode(attn.reshape(T, cfg.q_size), "o_proj"))
        if fuse8:
            w8g, swg, sg, ag = fp8["gate_up_proj"]
            x8 = ops.add_rmsnorm_fp8_(h, attn_delta, self.post_attention_layernorm,
                                      cfg.rms_eps, sg, ag)
            gu = ops.fp8_mm_prequant(x8, w8g, sg, swg)
            w8d, swd, sd, ad = fp8["down_proj"]
            m8 = ops.swiglu_fp8(gu, sd, ad)
            return self._tp_reduce(ops.fp8_mm_prequant(m8, w8d, sd, swd))
        x = ops.add_rmsnorm_(h, attn_delta, self.post_attention_layernorm, cfg.rms_eps)
        mlp_delta = self._tp_reduce(
            self._lin_decode(ops.swiglu(self._lin_decode(x, "gate_up_proj")), "down_proj"))
        return mlp_delta

    def _lin_decode(self, x: torch.Tensor, name: str) -> torch.Tensor:
        """Decode-path linear: hipBLASLt skinny GEMM (tuned algo; fp8 e4m3
        weights+activations when enable_fp8_decode is armed — rollout-only,
        the bf16 update corrects the distribution shift via TIS) + optional
        unmerged-LoRA term (rollouts normally run MERGED — merge_lora_ — so
        that branch is off; kept for logprob-exactness checks)."""
        fp8 = getattr(self, "_fp8", None)
        if fp8 is not None and name in fp8:
            w8, sw, scale_buf, amax_buf = fp8[name]
            y = ops.fp8_linear_delayed(x, w8, sw, scale_buf, amax_buf,
                                       w_bf16=getattr(self, name))
        else:
            y = ops.linear_decode(x, getattr(self, name))
        if self._lora_on and f"{name}_A" in self.lora:
            from rllm_amd.models import lora as _lora

            y = y + _lora.lora_delta(self, name, x)
        return y


class QwenModel(nn.Module):
    def __init__(self, cfg: ModelConfig, device: str = "cuda", max_positions: int = 65536):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Parameter(torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
        self.layers = nn.ModuleList([QwenLayer(cfg) for _ in range(cfg.num_layers)])
        self.norm = nn.Parameter(torch.ones(cfg.hidden_size, dtype=torch.bfloat16))
        if cfg.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = nn.Parameter(torch.empty(cfg.vocab_size, cfg.hidden_size, dtype=torch.bfloat16))
        self.gradient_checkpointing = False
        self._device = device
        self.to(device)
        cos_t, sin_t = ops.build_rope_tables(max_positions, cfg.head_dim, cfg.rope_theta, device) \
            if device != "meta" else (None, None)
        self.cos_t, self.sin_t = cos_t, sin_t

    @property
    def lm_weight(self) -> torch.Tensor:
        return self.embed_tokens if self.lm_head is None else self.lm_head

    def enable_fp8_decode(self) -> None:
        """(Re)quantize the four decode projections to OCP e4m3 with
        per-tensor scales. Call after every weight update (the engine's
        weight_version setter does) — the fp8 copies alias nothing."""
        names = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")
        fresh = not hasattr(self, "_fp8_scales")
        if fresh:
            n_sites = len(self.layers) * len(names)
            dev = next(self.parameters()).device
            # every site's scale/amax is a VIEW into these, so ONE kernel per
            # decode step folds all accumulated amaxes (fp8_scale_update_all)
            self._fp8_scales = torch.ones(n_sites, device=dev, dtype=torch.float32)
            self._fp8_amax = torch.zeros(n_sites, device=dev, dtype=torch.float32)
        with torch.no_grad():
            site = 0
            for layer in self.layers:
                old = getattr(layer, "_fp8", None)
                q = {}
                for name in names:
                    w8, sw = ops.fp8_quant(getattr(layer, name).detach())
                    if old is not None and name in old:
                        # keep buffer identity — captured graphs hold pointers
                        old_w8, old_sw, scale_v, amax_v = old[name]
                        old_w8.copy_(w8)
                        old_sw.copy_(sw)
                        q[name] = (old_w8, old_sw, scale_v, amax_v)
                    else:
                        q[name] = (w8, sw,
                                   self._fp8_scales[site : site + 1],
                                   self._fp8_amax[site : site + 1])
                    site += 1
                layer._fp8 = q
        self.fp8_decode = True

    def disable_fp8_decode(self) -> None:
        for layer in self.layers:
            if hasattr(layer, "_fp8"):
                del layer._fp8
        self.fp8_decode = False

    def _stamp_lora(self) -> None:
        """Propagate the model-level adapter state to the layers once per
        forward (merged adapters live inside the base weights, so the
        low-rank term must be off)."""
        on = bool(getattr(self, "lora_enabled", False)) and not getattr(self, "lora_merged", False)
        if on or getattr(self, "_lora_stamped", False):
            for layer in self.layers:
                layer._lora_on = on
            self._lora_stamped = on

    def init_random(self, seed: int = 0):
        """Random init at HF-like scales (std 0.02)."""
        gen = torch.Generator(device="cpu").manual_seed(seed)
        for name, p in self.named_parameters():
            if p.dim() == 2:
                init = torch.randn(p.shape, generator=gen, dtype=torch.float32) * 0.02
                with torch.no_grad():
                    p.copy_(init.to(p.dtype))
            elif "bias" in name:
                b = torch.randn(p.shape, generator=gen, dtype=torch.float32) * 0.02
                with torch.no_grad():
                    p.copy_(b.to(p.dtype))
            else:  # norms
                with torch.no_grad():
                    p.fill_(1.0)
        return self

    # -- training -------------------------------------------------------------
    use_flash_training_attention: bool = True
    sp_group = None  # Ulysses SP process group (parallel/ulysses.py); None = off

    def forward_train(self, input_ids: torch.Tensor, positions: torch.Tensor,
                      cu_seqlens: list[int]) -> torch.Tensor:
        """Packed varlen forward -> final hidden states [T, H] (after norm).

        Ulysses mode (self.sp_group set): input_ids/positions are this
        rank's token shard (parallel/ulysses.py shard_slice), cu_seqlens is
        the FULL packed batch; the returned hidden covers the local shard.
        """
        self._stamp_lora()
        hidden = self.embed_tokens[input_ids]
        flash_tiles = None
        # the MFMA training-attention kernel is specialized for D=128 (the
        # Qwen2.5/R1-Distill head size); other head dims (e.g. 0.5B's 64,
        # tiny test configs) take the chunked-matmul path
        if self.use_flash_training_attention and self.cfg.head_dim == 128:
            seqlens = [cu_seqlens[i + 1] - cu_seqlens[i] for i in range(len(cu_seqlens) - 1)]
            flash_tiles = make_prefill_tiles(seqlens, hidden.device)
        for layer in self.layers:
            if self.gradient_checkpointing and torch.is_grad_enabled():
                hidden = torch.utils.checkpoint.checkpoint(
                    layer.forward_train, hidden, positions, cu_seqlens,
                    self.cos_t, self.sin_t, flash_tiles, self.sp_group,
                    use_reentrant=False)
            else:
                hidden = layer.forward_train(hidden, positions, cu_seqlens,
                                             self.cos_t, self.sin_t, flash_tiles,
                                             self.sp_group)
        return ops.rmsnorm(hidden, self.norm, self.cfg.rms_eps)

    def logprobs_for_tokens(self, input_ids, positions, cu_seqlens, targets,
                            want_entropy: bool = True, temperature: float = 1.0):
        """Token logprob (+ entropy) for `targets` aligned with input rows.
        The caller masks out rows it does not need (prompt/observation rows
        get target -1 ... they still cost a gather, so callers typically
        index-select response rows first)."""
        hidden = self.forward_train(input_ids, positions, cu_seqlens)
        return ops.chunked_logprob(hidden, self.lm_weight, targets,
                                   temperature=temperature, want_entropy=want_entropy)

    # -- rollout --------------------------------------------------------------
    @torch.no_grad()
    def forward_prefill(self, input_ids, positions, tiles, kv_cache, slot_mapping):
        self._stamp_lora()
        hidden = self.embed_tokens[input_ids]
        for i, layer in enumerate(self.layers):
            hidden = layer.forward_prefill(hidden, positions, self.cos_t, self.sin_t,
                                           tiles, kv_cache, slot_mapping, i)
        return ops.rmsnorm(hidden, self.norm, self.cfg.rms_eps)

    @torch.no_grad()
    def forward_prefill_cached(self, input_ids, positions, tiles, kv_cache,
                               slot_suffix, slot_full, suffix_rows, T_full):
        """Suffix-only prefill against prefix-cached pages (engine prefix
        caching; see QwenLayer.forward_prefill_cached)."""
        self._stamp_lora()
        hidden = self.embed_tokens[input_ids]
        for i, layer in enumerate(self.layers):
            hidden = layer.forward_prefill_cached(hidden, positions, self.cos_t, self.sin_t,
                                                  tiles, kv_cache, slot_suffix, slot_full,
                                                  suffix_rows, T_full, i)
        return ops.rmsnorm(hidden, self.norm, self.cfg.rms_eps)

    @torch.no_grad()
    def forward_decode(self, input_ids, positions, kv_cache, slot_mapping, block_tables, seq_lens):
        self._stamp_lora()
        if getattr(self, "fp8_decode", False):
            # fold last step's activation amaxes into this step's scales
            ops.fp8_scale_update_all(self._fp8_scales, self._fp8_amax)
        h = self.embed_tokens[input_ids].contiguous()
        delta = None
        for i, layer in enumerate(self.layers):
            delta = layer.forward_decode_fused(h, delta, positions, self.cos_t, self.sin_t,
                                               kv_cache, slot_mapping, block_tables, seq_lens, i)
        return ops.add_rmsnorm_(h, delta, self.norm, self.cfg.rms_eps)

    @torch.no_grad()
    def logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if hidden.shape[0] <= 512:
            return ops.linear_decode(hidden, self.lm_weight)
        return torch.matmul(hidden, self.lm_weight.t())

    # -- HF checkpoint mapping --------------------------------------------------
    @torch.no_grad()
    def load_hf_state_dict(self, sd: dict[str, torch.Tensor]):
        """Map a HuggingFace Qwen2 state dict onto the fused layout."""
        cfg = self.cfg

        def cp(dst: torch.Tensor, key: str):
            dst.copy_(sd[key].to(dst.dtype))

        cp(self.embed_tokens, "model.embed_tokens.weight")
        cp(self.norm, "model.norm.weight")
        if self.lm_head is not None:
            cp(self.lm_head, "lm_head.weight")
        for i, layer in enumerate(self.layers):
            p = f"model.layers.{i}."
            cp(layer.input_layernorm, p + "input_layernorm.weight")
            cp(layer.post_attention_layernorm, p + "post_attention_layernorm.weight")
            q = sd[p + "self_attn.q_proj.weight"]
            k = sd[p + "self_attn.k_proj.weight"]
            v = sd[p + "self_attn.v_proj.weight"]
            layer.qkv_proj.copy_(torch.cat([q, k, v], dim=0).to(torch.bfloat16))
            if layer.qkv_bias is not None:
                qb = sd[p + "self_attn.q_proj.bias"]
                kb = sd[p + "self_attn.k_proj.bias"]
                vb = sd[p + "self_attn.v_proj.bias"]
                layer.qkv_bias.copy_(torch.cat([qb, kb, vb], dim=0).to(torch.bfloat16))
            cp(layer.o_proj, p + "self_attn.o_proj.weight")
            g = sd[p + "mlp.gate_proj.weight"]
            u = sd[p + "mlp.up_proj.weight"]
            layer.gate_up_proj.copy_(torch.cat([g, u], dim=0).to(torch.bfloat16))
            cp(layer.down_proj, p + "mlp.down_proj.weight")
        return self


def make_prefill_tiles(seqlens: list[int], device, tile: int = 64):
    """Host-side q-tile table for flash_prefill over packed varlen batches."""
    tile_seq_start, tile_row0, tile_seq_len = [], [], []
    start = 0
    for n in seqlens:
        for r0 in range(0, n, tile):
            tile_seq_start.append(start)
            tile_row0.append(start + r0)
            tile_seq_len.append(n)
        start += n
    mk = lambda x: torch.tensor(x, device=device, dtype=torch.int32)
    return mk(tile_seq_start), mk(tile_row0), mk(tile_seq_len)


def make_prefill_tiles_cached(full_lens: list[int], cached_lens: list[int], device, tile: int = 64):
    """Q-tile table covering only each sequence's SUFFIX (rows past the
    prefix-cached region), in FULL-length packed coordinates. The flash
    kernel's q_local0 = row0 - seq_start handles arbitrary suffix starts;
    its epilogue skips rows below the tile start."""
    tile_seq_start, tile_row0, tile_seq_len = [], [], []
    start = 0
    for n, c in zip(full_lens, cached_lens):
        for r0 in range(c, n, tile):
            tile_seq_start.append(start)
            tile_row0.append(start + r0)
            tile_seq_len.append(n)
        start += n
    mk = lambda x: torch.tensor(x, device=device, dtype=torch.int32)
    return mk(tile_seq_start), mk(tile_row0), mk(tile_seq_len)
