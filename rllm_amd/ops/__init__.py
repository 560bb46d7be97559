"""rllm_amd.ops — CDNA4 HIP kernel library + autograd wrappers.

The `_C` extension (built in-tree by rllm_amd/ops/build.py) is REQUIRED on
any machine with a GPU: ops fail loudly rather than falling back to eager
PyTorch, so a silently-degraded bench is impossible. On CPU-only machines
the module imports fine (for unit tests of the pure-Python layers) but any
kernel call raises.
"""

from __future__ import annotations

import importlib.util
from pathlib import Path

import torch

_C = None
_LOAD_ERROR: Exception | None = None


def _try_load():
    global _C, _LOAD_ERROR
    so = Path(__file__).parent / "_C.so"
    if not so.exists():
        _LOAD_ERROR = FileNotFoundError(
            f"{so} not found — build it with `python -m rllm_amd.ops.build` "
            f"(hipcc --offload-arch=gfx950)"
        )
        return
    try:
        spec = importlib.util.spec_from_file_location("rllm_amd.ops._C", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)  # type: ignore[union-attr]
        _C = mod
    except Exception as e:  # noqa: BLE001
        _LOAD_ERROR = e


_try_load()


def require_ext():
    """Return the _C extension or raise loudly (never a silent fallback)."""
    if _C is None:
        raise RuntimeError(
            f"rllm_amd HIP extension is not available: {_LOAD_ERROR}. "
            f"Refusing to fall back to eager PyTorch on the GPU path."
        ) from _LOAD_ERROR
    return _C


def ext_available() -> bool:
    return _C is not None


# ---------------------------------------------------------------------------
# Autograd wrappers
# ---------------------------------------------------------------------------


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, eps: float):
        C = require_ext()
        y, inv_rms = C.rmsnorm_fwd(x, weight, eps, True)
        ctx.save_for_backward(x, weight, inv_rms)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = require_ext()
        x, weight, inv_rms = ctx.saved_tensors
        dx, dw = C.rmsnorm_bwd(dy.contiguous(), x, weight, inv_rms)
        return dx, dw.to(weight.dtype), None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.requires_grad or weight.requires_grad:
        return _RMSNormFn.apply(x.contiguous(), weight, eps)
    C = require_ext()
    return C.rmsnorm_fwd(x.contiguous(), weight, eps, False)[0]


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gateup: torch.Tensor):
        C = require_ext()
        out = C.swiglu_fwd(gateup)
        ctx.save_for_backward(gateup)
        return out

    @staticmethod
    def backward(ctx, dout):
        C = require_ext()
        (gateup,) = ctx.saved_tensors
        return C.swiglu_bwd(dout.contiguous(), gateup)


def swiglu(gateup: torch.Tensor) -> torch.Tensor:
    """gateup [..., 2I] (gate | up) -> silu(gate) * up, [..., I]."""
    if gateup.requires_grad:
        return _SwiGLUFn.apply(gateup.contiguous())
    return require_ext().swiglu_fwd(gateup.contiguous())


class _RoPEFn(torch.autograd.Function):
    """Out-of-place autograd wrapper over the in-place rope kernel."""

    @staticmethod
    def forward(ctx, q, k, cos_tab, sin_tab, positions):
        C = require_ext()
        q = q.clone(memory_format=torch.contiguous_format)
        k = k.clone(memory_format=torch.contiguous_format)
        C.rope_inplace(q, k, cos_tab, sin_tab, positions, False)
        ctx.save_for_backward(cos_tab, sin_tab, positions)
        return q, k

    @staticmethod
    def backward(ctx, dq, dk):
        C = require_ext()
        cos_tab, sin_tab, positions = ctx.saved_tensors
        dq = dq.contiguous().clone()
        dk = dk.contiguous().clone()
        C.rope_inplace(dq, dk, cos_tab, sin_tab, positions, True)
        return dq, dk, None, None, None


def rope(q, k, cos_tab, sin_tab, positions):
    """Apply rotary embedding (never aliases its inputs).
    q [T,Hq,D], k [T,Hk,D], positions [T] int32."""
    if q.requires_grad or k.requires_grad:
        return _RoPEFn.apply(q, k, cos_tab, sin_tab, positions)
    C = require_ext()
    q = q.clone(memory_format=torch.contiguous_format)
    k = k.clone(memory_format=torch.contiguous_format)
    C.rope_inplace(q, k, cos_tab, sin_tab, positions, False)
    return q, k


def build_rope_tables(max_pos: int, head_dim: int, theta: float, device) -> tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed cos/sin tables [max_pos, head_dim/2] fp32 (guide: keep
    trig off the device for memory-bound RoPE)."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    pos = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(pos, inv_freq)
    return freqs.cos().float().to(device), freqs.sin().float().to(device)


class _ChunkedLogprobFn(torch.autograd.Function):
    """Fused token-logprob (+ entropy metric) over hidden @ lm_head^T without
    materializing [T, vocab] (K7/K12). GEMM chunks via hipBLASLt; online-LSE
    and dlogits formation are HIP kernels."""

    @staticmethod
    def forward(ctx, hidden: torch.Tensor, lm_weight: torch.Tensor, targets: torch.Tensor,
                chunk: int = 16384, temperature: float = 1.0, want_entropy: bool = True):
        C = require_ext()
        T = hidden.shape[0]
        V = lm_weight.shape[0]
        inv_temp = 1.0 / temperature
        m = torch.full((T,), -float("inf"), device=hidden.device, dtype=torch.float32)
        s = torch.zeros(T, device=hidden.device, dtype=torch.float32)
        e = torch.zeros(T, device=hidden.device, dtype=torch.float32) if want_entropy else None
        tgt_logit = torch.zeros(T, device=hidden.device, dtype=torch.float32)
        targets_i32 = targets.to(torch.int32)
        for c0 in range(0, V, chunk):
            c1 = min(V, c0 + chunk)
            logits = torch.matmul(hidden, lm_weight[c0:c1].t()).contiguous()
            C.lse_chunk_update(logits, m, s, e, tgt_logit, targets_i32, c0, inv_temp)
        lse = m + torch.log(s)
        logprob = tgt_logit - lse
        entropy = (lse - e / s) if want_entropy else None
        if entropy is not None:
            ctx.mark_non_differentiable(entropy)
        ctx.save_for_backward(hidden, lm_weight, targets_i32, lse)
        ctx.chunk = chunk
        ctx.inv_temp = inv_temp
        return logprob, entropy

    @staticmethod
    def backward(ctx, dlogprob, dentropy):
        C = require_ext()
        hidden, lm_weight, targets_i32, lse = ctx.saved_tensors
        V = lm_weight.shape[0]
        chunk = ctx.chunk
        dlp = dlogprob.contiguous().to(torch.float32)
        dhidden = torch.zeros_like(hidden)
        dweight = torch.zeros_like(lm_weight) if lm_weight.requires_grad else None
        for c0 in range(0, V, chunk):
            c1 = min(V, c0 + chunk)
            w_chunk = lm_weight[c0:c1]
            logits = torch.matmul(hidden, w_chunk.t()).contiguous()
            dlogits = C.ce_bwd_chunk(logits, lse, dlp, targets_i32, c0, ctx.inv_temp)
            dhidden += torch.matmul(dlogits, w_chunk)
            if dweight is not None:
                dweight[c0:c1] = torch.matmul(dlogits.t(), hidden)
        return dhidden, dweight, None, None, None, None


def chunked_logprob(hidden, lm_weight, targets, chunk: int = 16384,
                    temperature: float = 1.0, want_entropy: bool = True):
    """logprob[t] = log softmax(hidden[t] @ W^T / temp)[targets[t]].

    Returns (logprob [T] fp32, entropy [T] fp32 or None). Entropy is a
    metric only — no gradient flows through it.
    """
    return _ChunkedLogprobFn.apply(hidden, lm_weight, targets, chunk, temperature, want_entropy)


class _GRPOLossFn(torch.autograd.Function):
    """Per-token ratio-clip + KL loss; gradient computed in the same fused
    forward kernel (K8-K10)."""

    @staticmethod
    def forward(ctx, logprob, old_logprob, ref_logprob, advantages, tis_w,
                eps_lo: float, eps_hi: float, kl_beta: float):
        C = require_ext()
        loss_tok, dlp_tok, clipped = C.grpo_loss_fwd(
            logprob.contiguous(), old_logprob.contiguous(),
            ref_logprob.contiguous() if ref_logprob is not None else None,
            advantages.contiguous(),
            tis_w.contiguous() if tis_w is not None else None,
            eps_lo, eps_hi, kl_beta)
        ctx.save_for_backward(dlp_tok)
        ctx.mark_non_differentiable(clipped)
        return loss_tok, clipped

    @staticmethod
    def backward(ctx, dloss, dclipped):
        (dlp_tok,) = ctx.saved_tensors
        return dloss * dlp_tok, None, None, None, None, None, None, None


def grpo_loss_per_token(logprob, old_logprob, ref_logprob, advantages, tis_w=None,
                        eps_lo: float = 0.2, eps_hi: float | None = None, kl_beta: float = 0.0):
    """Returns (loss_tok [N], clipped [N]) — aggregate with masked mean per
    loss_agg_mode upstream."""
    if eps_hi is None:
        eps_hi = eps_lo
    return _GRPOLossFn.apply(logprob, old_logprob, ref_logprob, advantages, tis_w,
                             eps_lo, eps_hi, kl_beta)


# ---------------------------------------------------------------------------
# Raw kernel pass-throughs (inference path, no autograd)
# ---------------------------------------------------------------------------


def flash_prefill(q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale):
    return require_ext().flash_prefill(q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale)


class _FlashAttnFn(torch.autograd.Function):
    """Hand-written training flash attention (fwd emits LSE; bwd is the
    FA2-style two-pass recompute — attention.hip flash_bwd_*)."""

    @staticmethod
    def forward(ctx, q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale):
        C = require_ext()
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = C.flash_train_fwd(q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale)
        ctx.save_for_backward(q, k, v, o, lse, tile_seq_start, tile_row0, tile_seq_len)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        C = require_ext()
        q, k, v, o, lse, ts, tr, tl = ctx.saved_tensors
        dq, dk, dv = C.flash_train_bwd(q, k, v, o, dout.contiguous(), lse, ts, tr, tl, ctx.scale)
        return dq, dk, dv, None, None, None, None


def flash_attention_train(q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale):
    """Differentiable causal GQA flash attention (packed varlen)."""
    return _FlashAttnFn.apply(q, k, v, tile_seq_start, tile_row0, tile_seq_len, scale)


def paged_decode(q, k_pages, v_pages, block_tables, seq_lens, scale, n_splits: int = 0):
    return require_ext().paged_decode(q, k_pages, v_pages, block_tables, seq_lens, scale, n_splits)


def reshape_and_cache(k, v, k_pages, v_pages, slot_mapping):
    return require_ext().reshape_and_cache(k, v, k_pages, v_pages, slot_mapping)


def gather_cache(k_pages, v_pages, slot_mapping):
    """Inverse of reshape_and_cache: contiguous [T, Hk, D] K/V from pages."""
    return require_ext().gather_cache(k_pages, v_pages, slot_mapping)


def qkv_rope_cache(qkv, bias, k_pages, v_pages, cos_tab, sin_tab, positions, slot_mapping, Hq, Hk):
    """Fused bias + rope + paged KV write + contiguous-q extract (rollout path)."""
    return require_ext().qkv_rope_cache(qkv, bias, k_pages, v_pages, cos_tab, sin_tab,
                                        positions, slot_mapping, Hq, Hk)


def add_rmsnorm_(h, delta, weight, eps: float = 1e-6):
    """h += delta (in place); returns rmsnorm(h) * weight. delta may be None."""
    return require_ext().add_rmsnorm_(h, delta, weight, eps)


def skinny_gemm(a, w, bias=None):
    """Weight-streaming GEMM for M<=512 (decode): C = a @ w.T (+ bias)."""
    return require_ext().skinny_gemm(a, w, bias)


import os as _os

_USE_SKINNY = _os.environ.get("RLLM_SKINNY_GEMM", "0") == "1"


def fp8_quant(t: torch.Tensor):
    """Per-tensor OCP e4m3 quantization -> (t8, scale [1] fp32 on device);
    original ≈ t8.float() * scale. Graph-capturable (amax/div/cast only)."""
    scale = (t.float().abs().amax() / 448.0).clamp(min=1e-8).reshape(1)
    t8 = (t.float() / scale).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return t8.contiguous(), scale


def fp8_linear(x: torch.Tensor, w8: torch.Tensor, sw: torch.Tensor,
               w_bf16: torch.Tensor | None = None):
    """y = x @ dequant(w8)^T with dynamic per-call activation quantization
    (torch-level; ~6 small kernels — prefer fp8_linear_delayed on hot
    paths). Untuned shapes fall back to the bf16 weight when provided."""
    C = require_ext()
    M, K = x.shape[0], x.shape[-1]
    N = w8.shape[0]
    if not C.hbl_fp8_has(M, N, K):
        if w_bf16 is not None:
            return linear_decode(x, w_bf16)
        raise RuntimeError(f"fp8 shape {(M, N, K)} not tuned")
    x8, sx = fp8_quant(x)
    return C.hbl_fp8_mm(x8, w8, sx, sw)


def fp8_linear_delayed(x: torch.Tensor, w8: torch.Tensor, sw: torch.Tensor,
                       scale_buf: torch.Tensor, amax_buf: torch.Tensor,
                       w_bf16: torch.Tensor | None = None,
                       update_scale: bool = False):
    """Hot-path fp8 linear with DELAYED activation scaling: quantize with
    this site's PREVIOUS scale while accumulating the new amax in the same
    (native cvt_pk_fp8) kernel. Two launches (quant, GEMM); the scale fold
    happens once per step via `fp8_scale_update_all` over every site's
    view into a shared buffer (pass update_scale=True for standalone use).
    Graph-capture safe — scale/amax are persistent per-site buffers."""
    C = require_ext()
    M, K = x.shape[0], x.shape[-1]
    N = w8.shape[0]
    if not C.hbl_fp8_has(M, N, K):
        if w_bf16 is not None:
            return linear_decode(x, w_bf16)
        raise RuntimeError(f"fp8 shape {(M, N, K)} not tuned")
    x8 = torch.empty(x.shape, device=x.device, dtype=torch.float8_e4m3fn)
    C.fp8_quant_delayed(x.contiguous(), x8, scale_buf, amax_buf)
    y = C.hbl_fp8_mm(x8, w8, scale_buf, sw)
    if update_scale:
        C.fp8_scale_update(scale_buf, amax_buf)
    return y


def add_rmsnorm_fp8_(h, delta, weight, eps, scale, amax):
    """Fused residual add (in place on h) + RMSNorm emitting e4m3 with
    delayed scaling — the producing kernel quantizes the next GEMM's input
    directly (no separate quant launch, no bf16 round trip)."""
    return require_ext().add_rmsnorm_fp8_(h, delta, weight, eps, scale, amax)


def swiglu_fp8(gateup, scale, amax):
    """SwiGLU emitting e4m3 with delayed scaling (input to the down GEMM)."""
    return require_ext().swiglu_fp8(gateup, scale, amax)


def fp8_mm_prequant(x8, w8, sx, sw):
    """Tuned fp8 GEMM over a PRE-quantized activation (produced by
    add_rmsnorm_fp8_/swiglu_fp8); bf16 out."""
    return require_ext().hbl_fp8_mm(x8, w8, sx, sw)


def fp8_scale_update_all(scales: torch.Tensor, amaxes: torch.Tensor):
    """Fold every call site's accumulated amax into its scale (one launch;
    called at the top of each decode step)."""
    require_ext().fp8_scale_update(scales, amaxes)


_FP8_TUNED: set = set()


def pretune_fp8_decode_shapes(shapes, device="cuda", iters: int = 20, verbose: bool = True):
    """fp8 analogue of pretune_decode_shapes. Returns {shape: best_us};
    shapes that produce no valid fp8 algo are skipped (bf16 fallback)."""
    if _TUNE_DISABLED:
        return {}
    C = require_ext()
    out = {}
    for (M, N, K) in shapes:
        key = (int(M), int(N), int(K))
        if key in _FP8_TUNED:
            continue
        x8, sx = fp8_quant(torch.randn(M, K, device=device))
        w8, sw = fp8_quant(torch.randn(N, K, device=device))
        try:
            best_us, n_cand, n_valid, idx = C.hbl_fp8_tune(x8, w8, sx, sw, iters)
        except RuntimeError as e:
            if verbose:
                print(f"[ops.pretune-fp8] {key}: no valid algo ({e}); bf16 fallback")
            continue
        _FP8_TUNED.add(key)
        out[key] = best_us
        if verbose:
            print(f"[ops.pretune-fp8] {key}: {best_us:.2f} us "
                  f"(algo {int(idx)}, {int(n_valid)}/{int(n_cand)} valid)")
    return out


def skinny_gemm_v3(x, w, bias=None):
    """v3 hand-written skinny GEMM (64x64 wave tiles, double-buffered LDS,
    partials split-K). Experimental: enable per-shape only where it beats
    the tuned hipBLASLt path (see profiles/)."""
    return require_ext().skinny_gemm_v3(x, w, bias)


def linear_decode(x, w, bias=None):
    """Decode-path linear. Default: hipBLASLt with per-shape TUNED algo
    selection (hbl_tuned.hip) when `pretune_decode_shapes` has run for this
    (M,N,K) — torch.matmul's default heuristic leaves a ~19-25 us/call
    floor at M<=512 on MI355X. Falls back to the default heuristic for
    untuned shapes. The in-repo weight-streaming skinny kernel is correct
    but slower (RLLM_SKINNY_GEMM=1 to opt in; redesign notes in TODO)."""
    if _USE_SKINNY and x.shape[0] <= 512 and x.shape[-1] % 32 == 0:
        return require_ext().skinny_gemm(x, w, bias)
    if bias is None and _TUNED_SHAPES and (x.shape[0], w.shape[0], w.shape[1]) in _TUNED_SHAPES:
        return require_ext().hbl_mm(x, w)
    return torch.nn.functional.linear(x, w, bias)


_TUNED_SHAPES: set = set()
_TUNE_DISABLED = _os.environ.get("RLLM_TUNED_GEMM", "1") == "0"


def pretune_decode_shapes(shapes, device="cuda", iters: int = 20, verbose: bool = True):
    """Sweep hipblaslt's heuristic candidates once per (M, N, K) decode
    shape and cache the fastest valid algo (hbl_tuned.hip). Call from
    engine init BEFORE any hipGraph capture; linear_decode then uses the
    tuned algo for exactly these shapes. Returns {shape: best_us}."""
    if _TUNE_DISABLED:
        return {}
    C = require_ext()
    out = {}
    for (M, N, K) in shapes:
        key = (int(M), int(N), int(K))
        if key in _TUNED_SHAPES:
            continue
        x = torch.randn(M, K, device=device).to(torch.bfloat16)
        w = torch.randn(N, K, device=device).to(torch.bfloat16)
        try:
            best_us, n_cand, n_valid, idx = C.hbl_tune(x, w, iters)
        except RuntimeError as e:  # no valid algo — keep default path
            if verbose:
                print(f"[ops.pretune] {key}: tune failed ({e}); using default heuristic")
            continue
        _TUNED_SHAPES.add(key)
        out[key] = best_us
        if verbose:
            print(f"[ops.pretune] {key}: {best_us:.2f} us "
                  f"(algo {int(idx)}, {int(n_valid)}/{int(n_cand)} valid)")
    return out


def decode_gemm_shapes(cfg, batch_sizes) -> list:
    """The (M, N, K) set the fused decode path hits for a model config:
    qkv, o, gate_up, down projections (+ lm head for logits) at each padded
    graph batch size."""
    shapes = []
    for M in batch_sizes:
        shapes += [
            (M, cfg.q_size + 2 * cfg.kv_size, cfg.hidden_size),
            (M, cfg.hidden_size, cfg.q_size),
            (M, 2 * cfg.intermediate_size, cfg.hidden_size),
            (M, cfg.hidden_size, cfg.intermediate_size),
            (M, cfg.vocab_size, cfg.hidden_size),
        ]
    return shapes


def sample_logprob(logits, temperature: float, seed: int, step: int, step_tensor=None):
    return require_ext().sample_logprob(logits, temperature, seed, step, step_tensor)


def gather_logprob(logits, tokens, temperature: float = 1.0):
    return require_ext().gather_logprob(logits, tokens, temperature)


def adamw_step(grad, master, m, v, param, gnorm_sq, *, lr, beta1=0.9, beta2=0.999,
               eps=1e-8, weight_decay=0.0, step, grad_clip=0.0, grad_scale=1.0):
    require_ext().adamw_step(grad, master, m, v, param, gnorm_sq, lr, beta1, beta2,
                             eps, weight_decay, step, grad_clip, grad_scale)


def grad_sq_sum(grad, grad_scale: float = 1.0):
    return require_ext().grad_sq_sum(grad, grad_scale)
