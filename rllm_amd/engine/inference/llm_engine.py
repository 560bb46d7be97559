"""Continuous-batching LLM engine over the paged KV cache
(K1-K6 in SURVEY.md §2.E — the vLLM replacement behind the gateway).

Design (MI355X-first):
* One engine per GPU; model weights are SHARED TENSORS with the trainer's
  actor on the same device, so weight sync in colocated mode is free
  (288 GB HBM: actor + ref + KV co-resident; no sleep/offload gymnastics).
* step() = one scheduler iteration: admit waiting requests into a prefill
  batch bounded by max_num_batched_tokens (reference knob yaml:86), else
  one decode step over all running sequences (batch cap max_num_seqs).
* Per-token logprob of the sampled token captured by the fused
  sample_logprob kernel, 1:1 with completion ids (types.py:167-168).
* abort/resume supported for partial rollout (async_agent_loop.py:11-14).
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from enum import Enum
from typing import Callable

import torch

from rllm_amd import ops
from rllm_amd.engine.inference.kv_cache import PAGE_SIZE, KVCache, PrefixCache
from rllm_amd.models.qwen import QwenModel, make_prefill_tiles, make_prefill_tiles_cached


@dataclass
class SamplingParams:
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = -1
    max_tokens: int = 512
    stop_token_ids: list[int] = field(default_factory=list)
    logprobs: bool = True
    seed: int = 0

    @classmethod
    def from_dict(cls, d: dict) -> "SamplingParams":
        return cls(
            temperature=float(d.get("temperature", 1.0)),
            top_p=float(d.get("top_p", 1.0)),
            top_k=int(d.get("top_k", -1)),
            max_tokens=int(d.get("max_tokens", d.get("max_completion_tokens", 512))),
            stop_token_ids=list(d.get("stop_token_ids", [])),
            logprobs=bool(d.get("logprobs", True)),
            seed=int(d.get("seed", 0)),
        )


class SeqState(Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"
    ABORTED = "aborted"


class Sequence:
    __slots__ = ("request_id", "prompt_ids", "output_ids", "logprobs", "params",
                 "state", "pages", "finish_reason", "weight_version", "arrival_time",
                 "first_token_time", "step_counter", "cached_len", "n_cached_pages")

    def __init__(self, request_id: str, prompt_ids: list[int], params: SamplingParams,
                 weight_version: int = 0):
        self.request_id = request_id
        self.prompt_ids = list(prompt_ids)
        self.output_ids: list[int] = []
        self.logprobs: list[float] = []
        self.params = params
        self.state = SeqState.WAITING
        self.pages: list[int] = []
        self.finish_reason: str | None = None
        self.weight_version = weight_version
        self.arrival_time = time.monotonic()
        self.first_token_time: float | None = None
        self.step_counter = 0
        self.cached_len = 0        # prefix-cached tokens at prefill admission
        self.n_cached_pages = 0    # leading shared pages in self.pages

    @property
    def total_len(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)

    @property
    def last_token(self) -> int:
        return self.output_ids[-1] if self.output_ids else self.prompt_ids[-1]


@dataclass
class RequestOutput:
    request_id: str
    prompt_ids: list[int]
    token_ids: list[int]
    logprobs: list[float]
    finish_reason: str
    weight_version: int
    metrics: dict[str, float] = field(default_factory=dict)


class LLMEngine:
    """Synchronous engine core. Drive with step(); or use generate() for a
    blocking batch. The async server wraps this in a background thread."""

    def __init__(self, model: QwenModel, kv_cache: KVCache | None = None,
                 max_num_seqs: int = 1024, max_num_batched_tokens: int = 8192,
                 kv_budget_bytes: int | None = None, eos_token_id: int | None = None,
                 seed: int = 0, use_hip_graph: bool = True, max_model_len: int = 8192,
                 enable_prefix_caching: bool = True, fp8_decode: bool | None = None):
        self.model = model
        self.cfg = model.cfg
        self.device = next(model.parameters()).device
        if kv_cache is None:
            if kv_budget_bytes is None:
                free, _total = torch.cuda.mem_get_info(self.device)
                kv_budget_bytes = int(free * 0.6)
            kv_cache = KVCache.from_memory_budget(
                self.cfg.num_layers, self.cfg.num_kv_heads, self.cfg.head_dim,
                kv_budget_bytes, device=str(self.device))
        self.kv = kv_cache
        self.max_num_seqs = max_num_seqs
        self.max_num_batched_tokens = max_num_batched_tokens
        self.eos_token_id = eos_token_id
        self.seed = seed
        self._weight_version = 0
        self._paused = False
        # automatic prefix caching: cross-request page reuse keyed on token
        # content (multi-turn/cumulative rollouts share prompt prefixes)
        self.prefix_cache = PrefixCache(self.kv) if enable_prefix_caching else None
        # fp8 rollout (opt-in, RLLM_FP8_ROLLOUT=1): e4m3 decode GEMMs;
        # weights re-quantized on every weight-version bump
        if fp8_decode is None:
            import os as _os

            fp8_decode = _os.environ.get("RLLM_FP8_ROLLOUT", "0") == "1"
        self.fp8_decode = fp8_decode
        if fp8_decode:
            self.model.enable_fp8_decode()

        self.waiting: list[Sequence] = []
        self.running: list[Sequence] = []
        self.finished: dict[str, Sequence] = {}
        self._sample_step = 0
        # batching diagnostics (perf scripts read these)
        self.stats = {"decode_calls": 0, "decode_batch_sum": 0, "decode_tokens": 0,
                      "prefill_calls": 0, "prefill_tokens": 0}

        # ---- hipGraph decode state (launch-bound decode: ~300 dispatches /
        # token collapse into one graph replay; guide §1 "capture
        # launch-bound inner loops in hipGraphs") ----
        self.use_hip_graph = use_hip_graph
        self.max_model_len = max_model_len
        self._bt_width = (max_model_len + PAGE_SIZE - 1) // PAGE_SIZE
        self._graphs: dict = {}        # (Bpad, temp) -> (graph, tokens_out, lps_out)
        self._graph_pool = None
        self._gb = None                # static device buffers dict
        self._hb = None                # pinned host staging dict
        # ---- hipGraph prefill state: the training workloads prefill in
        # UNIFORM chunks (max_num_batched_tokens), so a handful of exact
        # (T, n_tiles) keys cover every call; anything else runs eager ----
        self._prefill_graphs: dict = {}  # (T, n_tiles) -> dict(graph, dev, host, hidden)
        self._prefill_graph_cap = 8

    # ------------------------------------------------------------------
    @property
    def weight_version(self) -> int:
        return self._weight_version

    @weight_version.setter
    def weight_version(self, v: int) -> None:
        if v != self._weight_version:
            if self.prefix_cache is not None:
                # cached KV was computed under the old weights — stale
                self.prefix_cache.clear()
            if self.fp8_decode:
                self.model.enable_fp8_decode()  # re-quantize fresh weights
        self._weight_version = v

    def _alloc_pages(self, n: int) -> list[int]:
        """Page allocation that evicts idle prefix-cache pages under pressure."""
        short = n - self.kv.num_free_pages
        if short > 0 and self.prefix_cache is not None:
            self.prefix_cache.evict(short)
        return self.kv.alloc(n)

    def _release_pages(self, pages: list[int]) -> None:
        if self.prefix_cache is not None:
            self.prefix_cache.release(pages)
        else:
            self.kv.free(pages)

    def _effective_free_pages(self) -> int:
        extra = self.prefix_cache.evictable if self.prefix_cache is not None else 0
        return self.kv.num_free_pages + extra

    # ------------------------------------------------------------------
    # Request lifecycle
    # ------------------------------------------------------------------
    def add_request(self, request_id: str, prompt_ids: list[int], params: SamplingParams) -> None:
        seq = Sequence(request_id, prompt_ids, params, weight_version=self.weight_version)
        if len(prompt_ids) >= self.max_model_len:
            # MAX_PROMPT_LENGTH_EXCEEDED: fail loudly instead of letting the
            # sequence outgrow the block-table width mid-decode
            self._finish(seq, "error")
            return
        self.waiting.append(seq)

    def abort(self, request_id: str) -> None:
        for lst in (self.waiting, self.running):
            for seq in lst:
                if seq.request_id == request_id:
                    lst.remove(seq)
                    self._finish(seq, "abort")
                    return

    def has_unfinished(self) -> bool:
        return bool(self.waiting or self.running)

    def pause(self):
        """Stop admitting new prefills (weight-sync drain protocol)."""
        self._paused = True

    def resume(self):
        self._paused = False

    def pop_finished(self) -> list[RequestOutput]:
        outs = []
        for seq in self.finished.values():
            outs.append(RequestOutput(
                request_id=seq.request_id,
                prompt_ids=seq.prompt_ids,
                token_ids=seq.output_ids,
                logprobs=seq.logprobs,
                finish_reason=seq.finish_reason or "stop",
                weight_version=seq.weight_version,
                metrics={"ttft_s": (seq.first_token_time - seq.arrival_time) if seq.first_token_time else 0.0},
            ))
        self.finished.clear()
        return outs

    # ------------------------------------------------------------------
    # Scheduler step
    # ------------------------------------------------------------------
    def step(self) -> int:
        """One iteration. Returns number of tokens processed."""
        batch = self._schedule_prefill()
        if batch:
            return self._run_prefill(batch)
        if self.running:
            return self._run_decode()
        return 0

    def _schedule_prefill(self) -> list[Sequence]:
        if self._paused or not self.waiting:
            return []
        batch: list[Sequence] = []
        tokens = 0
        pc = self.prefix_cache
        while self.waiting and len(self.running) + len(batch) < self.max_num_seqs:
            seq = self.waiting[0]
            n = seq.total_len
            need = KVCache.pages_needed(n + 1)
            if need > self.kv.num_pages - 1:
                # can NEVER fit, even with the whole pool: fail loudly
                # instead of spinning (liveness guard)
                self.waiting.pop(0)
                self._finish(seq, "error")
                continue
            # prefix cache: reuse pages whose token content matches
            matched: list[int] = []
            if pc is not None:
                matched = pc.match(seq.prompt_ids + seq.output_ids)
            # always leave >=1 token to prefill (we need its logits)
            cached_len = min(PAGE_SIZE * len(matched), n - 1)
            suffix = n - cached_len
            if batch and tokens + suffix > self.max_num_batched_tokens:
                if matched:
                    pc.release(matched)
                break
            if need - len(matched) > self._effective_free_pages():
                if matched:
                    pc.release(matched)
                break
            seq.pages = matched + self._alloc_pages(need - len(matched))
            seq.cached_len = cached_len
            seq.n_cached_pages = len(matched)
            batch.append(self.waiting.pop(0))
            tokens += suffix
        return batch

    def _slot(self, seq: Sequence, pos: int) -> int:
        return seq.pages[pos // PAGE_SIZE] * PAGE_SIZE + pos % PAGE_SIZE

    def _run_prefill(self, batch: list[Sequence]) -> int:
        if any(seq.cached_len > 0 for seq in batch):
            n_tok = self._run_prefill_cached(batch)
        else:
            n_tok = self._run_prefill_full(batch)
        self.stats["prefill_calls"] += 1
        self.stats["prefill_tokens"] += n_tok
        for seq in batch:
            if seq.state != SeqState.FINISHED:
                seq.state = SeqState.RUNNING
                self.running.append(seq)
            if seq.first_token_time is None:
                seq.first_token_time = time.monotonic()
        return n_tok

    def _run_prefill_full(self, batch: list[Sequence]) -> int:
        device = self.device
        input_ids, positions, slot_mapping, seqlens, last_rows = [], [], [], [], []
        row = 0
        for seq in batch:
            toks = seq.prompt_ids + seq.output_ids  # resume-aware (preemption)
            n = len(toks)
            input_ids.extend(toks)
            positions.extend(range(n))
            slot_mapping.extend(self._slot(seq, p) for p in range(n))
            seqlens.append(n)
            row += n
            last_rows.append(row - 1)

        T = len(input_ids)
        tile = 64
        n_tiles = sum((n + tile - 1) // tile for n in seqlens)
        key = (T, n_tiles)
        if self.use_hip_graph and (key in self._prefill_graphs
                                   or len(self._prefill_graphs) < self._prefill_graph_cap):
            hidden = self._run_prefill_graph(key, input_ids, positions, slot_mapping, seqlens)
        else:
            ids_t = torch.tensor(input_ids, device=device, dtype=torch.long)
            pos_t = torch.tensor(positions, device=device, dtype=torch.int32)
            slots_t = torch.tensor(slot_mapping, device=device, dtype=torch.int32)
            tiles = make_prefill_tiles(seqlens, device)
            hidden = self.model.forward_prefill(ids_t, pos_t, tiles, self.kv, slots_t)
        last_hidden = hidden[torch.tensor(last_rows, device=device, dtype=torch.long)]
        self._sample_and_append(batch, last_hidden)
        return len(input_ids)

    def _run_prefill_graph(self, key: tuple, input_ids, positions, slot_mapping, seqlens):
        """Replay (or capture) the full-prefill forward for this exact
        (token count, tile count) shape. Inputs stage through pinned host
        buffers into the captured static device buffers; sampling and the
        last-row gather stay eager (per-seq params). ~250 launches/call
        collapse into one replay (TODO r1 #6: ~0.4 s/step of prefill launch
        overhead at the flagship config)."""
        T, n_tiles = key
        dev = self.device
        entry = self._prefill_graphs.get(key)
        if entry is None:
            dbuf = {
                "ids": torch.zeros(T, device=dev, dtype=torch.long),
                "pos": torch.zeros(T, device=dev, dtype=torch.int32),
                "slot": torch.zeros(T, device=dev, dtype=torch.int32),
                "tss": torch.zeros(n_tiles, device=dev, dtype=torch.int32),
                "tr0": torch.zeros(n_tiles, device=dev, dtype=torch.int32),
                "tsl": torch.zeros(n_tiles, device=dev, dtype=torch.int32),
            }
            hbuf = {k: torch.zeros(v.shape, dtype=v.dtype, pin_memory=True)
                    for k, v in dbuf.items()}
            entry = {"dev": dbuf, "host": hbuf, "graph": None, "hidden": None}
            self._prefill_graphs[key] = entry
        dbuf, hbuf = entry["dev"], entry["host"]

        # stage host → pinned → device
        hbuf["ids"].numpy()[:] = input_ids
        hbuf["pos"].numpy()[:] = positions
        hbuf["slot"].numpy()[:] = slot_mapping
        tss, tr0, tsl = [], [], []
        start = 0
        for n in seqlens:
            for r0 in range(0, n, 64):
                tss.append(start)
                tr0.append(start + r0)
                tsl.append(n)
            start += n
        hbuf["tss"].numpy()[:] = tss
        hbuf["tr0"].numpy()[:] = tr0
        hbuf["tsl"].numpy()[:] = tsl
        for k in dbuf:
            dbuf[k].copy_(hbuf[k], non_blocking=True)

        def compute():
            return self.model.forward_prefill(
                dbuf["ids"], dbuf["pos"], (dbuf["tss"], dbuf["tr0"], dbuf["tsl"]),
                self.kv, dbuf["slot"])

        if entry["graph"] is None:
            # warm up eager once (workspace/algo init), then capture
            torch.cuda.synchronize()
            compute()
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            if self._graph_pool is None:
                with torch.cuda.graph(graph):
                    hidden = compute()
                self._graph_pool = graph.pool()
            else:
                with torch.cuda.graph(graph, pool=self._graph_pool):
                    hidden = compute()
            entry["graph"], entry["hidden"] = graph, hidden
        # capture does NOT execute the kernels — always replay
        entry["graph"].replay()
        return entry["hidden"]

    def _run_prefill_cached(self, batch: list[Sequence]) -> int:
        """Suffix-only prefill: cached prefix K/V come from shared pages;
        only tokens past each sequence's cached_len run the forward."""
        device = self.device
        input_ids, positions, slot_suffix, slot_full = [], [], [], []
        full_lens, cached_lens, suffix_rows, last_rows = [], [], [], []
        full_row = 0
        suffix_row = 0
        for seq in batch:
            toks = seq.prompt_ids + seq.output_ids
            n, c = len(toks), seq.cached_len
            input_ids.extend(toks[c:])
            positions.extend(range(c, n))
            slot_suffix.extend(self._slot(seq, p) for p in range(c, n))
            slot_full.extend(self._slot(seq, p) for p in range(n))
            suffix_rows.extend(range(full_row + c, full_row + n))
            full_lens.append(n)
            cached_lens.append(c)
            full_row += n
            suffix_row += n - c
            last_rows.append(suffix_row - 1)

        ids_t = torch.tensor(input_ids, device=device, dtype=torch.long)
        pos_t = torch.tensor(positions, device=device, dtype=torch.int32)
        slot_s = torch.tensor(slot_suffix, device=device, dtype=torch.int32)
        slot_f = torch.tensor(slot_full, device=device, dtype=torch.int32)
        rows_t = torch.tensor(suffix_rows, device=device, dtype=torch.long)
        tiles = make_prefill_tiles_cached(full_lens, cached_lens, device)

        hidden = self.model.forward_prefill_cached(
            ids_t, pos_t, tiles, self.kv, slot_s, slot_f, rows_t, full_row)
        last_hidden = hidden[torch.tensor(last_rows, device=device, dtype=torch.long)]
        self._sample_and_append(batch, last_hidden)
        return len(input_ids)

    def _preempt(self, seq: Sequence) -> None:
        """KV pressure: recycle the sequence's pages and requeue it. On
        re-admission the prefill recomputes KV for prompt+generated-so-far
        and decoding continues — the same mechanism gives partial-rollout
        resume (reference async_agent_loop.py:11-14)."""
        if seq in self.running:
            self.running.remove(seq)
        if seq.pages:
            self._release_pages(seq.pages)
            seq.pages = []
        seq.cached_len = 0
        seq.n_cached_pages = 0
        seq.state = SeqState.WAITING
        self.waiting.insert(0, seq)
        logger_warning = getattr(self, "_preempt_count", 0) + 1
        self._preempt_count = logger_warning

    def _grow_pages(self, seq: Sequence, pos: int) -> bool:
        """Ensure capacity for position pos; False if the pool is exhausted
        even after preempting younger sequences."""
        while KVCache.pages_needed(pos + 1) > len(seq.pages):
            try:
                seq.pages.extend(self._alloc_pages(1))
            except MemoryError:
                victim = None
                for cand in reversed(self.running):
                    if cand is not seq:
                        victim = cand
                        break
                if victim is None:
                    self._preempt(seq)
                    return False
                self._preempt(victim)
        return True

    def _run_decode(self) -> int:
        batch = list(self.running)  # snapshot: preemption mutates self.running
        temp = batch[0].params.temperature
        uniform = all(s.params.temperature == temp and s.params.top_k <= 0 and s.params.top_p >= 1.0
                      for s in batch)
        fits = all(s.total_len + 1 < self.max_model_len for s in batch)
        if self.use_hip_graph and uniform and fits and len(batch) <= 1024:
            n = self._run_decode_graph(batch, temp)
        else:
            n = self._run_decode_eager(batch)
        self.stats["decode_calls"] += 1
        self.stats["decode_batch_sum"] += len(batch)
        self.stats["decode_tokens"] += n
        self.running = [s for s in batch if s.state == SeqState.RUNNING]
        return n

    def _run_decode_eager(self, batch: list[Sequence]) -> int:
        device = self.device
        active: list[Sequence] = []
        input_ids, positions, slot_mapping, seq_lens = [], [], [], []
        for seq in batch:
            if seq.state != SeqState.RUNNING:
                continue  # preempted by an earlier seq's page grab
            pos = seq.total_len  # position of the NEW token
            if not self._grow_pages(seq, pos):
                continue
            active.append(seq)
            input_ids.append(seq.last_token)
            positions.append(pos)
            slot_mapping.append(self._slot(seq, pos))
            seq_lens.append(pos + 1)
        if not active or any(s.state != SeqState.RUNNING for s in active):
            return 0  # late preemption invalidated an earlier-staged seq
        batch = active
        B = len(batch)
        max_pages = max(len(s.pages) for s in batch)
        block_tables = torch.zeros(B, max_pages, device=device, dtype=torch.int32)
        for i, seq in enumerate(batch):
            block_tables[i, : len(seq.pages)] = torch.tensor(seq.pages, dtype=torch.int32)

        ids_t = torch.tensor(input_ids, device=device, dtype=torch.long)
        pos_t = torch.tensor(positions, device=device, dtype=torch.int32)
        slots_t = torch.tensor(slot_mapping, device=device, dtype=torch.int32)
        lens_t = torch.tensor(seq_lens, device=device, dtype=torch.int32)

        hidden = self.model.forward_decode(ids_t, pos_t, self.kv, slots_t, block_tables, lens_t)
        self._sample_and_append(batch, hidden)
        return B

    # ------------------------------------------------------------------
    # hipGraph decode path
    # ------------------------------------------------------------------
    _GRAPH_SIZES = [1, 2, 4, 8, 16, 32, 48, 64, 96, 128, 192, 256, 384, 512, 768, 1024]

    #: max decode steps per graph block (in-graph state advance between
    #: replays: no host work, no H2D staging, no D2H sync inside a block)
    BLOCK_MAX = 16
    #: block length when stop tokens are live — bounds over-decode past EOS
    EOS_BLOCK = 4

    def _init_graph_buffers(self):
        dev = self.device
        Bmax = min(self.max_num_seqs, 1024)
        K = self.BLOCK_MAX
        self._gb = {
            "tok": torch.zeros(Bmax, device=dev, dtype=torch.long),
            "pos": torch.zeros(Bmax, device=dev, dtype=torch.int32),
            "slot": torch.zeros(Bmax, device=dev, dtype=torch.int32),
            "bt": torch.zeros(Bmax, self._bt_width, device=dev, dtype=torch.int32),
            "len": torch.ones(Bmax, device=dev, dtype=torch.int32),
            "step": torch.zeros(1, device=dev, dtype=torch.int32),
            # block machinery: per-step slot schedule, in-block index, and
            # token/logprob history read back ONCE per block
            "slot_sched": torch.zeros(K, Bmax, device=dev, dtype=torch.int32),
            "blk_i": torch.zeros(1, device=dev, dtype=torch.long),
            # int32: the fused sampler emits int32 tokens and index_copy_
            # requires matching dtypes
            "tok_hist": torch.zeros(K, Bmax, device=dev, dtype=torch.int32),
            "lp_hist": torch.zeros(K, Bmax, device=dev, dtype=torch.float32),
        }
        self._hb = {
            "tok": torch.zeros(Bmax, dtype=torch.long, pin_memory=True),
            "pos": torch.zeros(Bmax, dtype=torch.int32, pin_memory=True),
            "slot_sched": torch.zeros(K, Bmax, dtype=torch.int32, pin_memory=True),
            "bt": torch.zeros(Bmax, self._bt_width, dtype=torch.int32, pin_memory=True),
            "len": torch.ones(Bmax, dtype=torch.int32, pin_memory=True),
        }

    def _decode_compute(self, Bp: int, temp: float):
        """One decode step INCLUDING the state advance for the next step —
        a block of R steps is R back-to-back replays with zero host work."""
        g = self._gb
        g["step"].add_(1)
        hidden = self.model.forward_decode(g["tok"][:Bp], g["pos"][:Bp], self.kv,
                                           g["slot"][:Bp], g["bt"][:Bp], g["len"][:Bp])
        logits = self.model.logits(hidden).contiguous()
        tokens, lps = ops.sample_logprob(logits, temp, self.seed, 0, step_tensor=g["step"])
        i = g["blk_i"]
        g["tok_hist"][:, :Bp].index_copy_(0, i, tokens.unsqueeze(0))
        g["lp_hist"][:, :Bp].index_copy_(0, i, lps.unsqueeze(0))
        # advance: next step's inputs (sampled token, pos+1, next slot row)
        g["tok"][:Bp].copy_(tokens)
        g["pos"][:Bp].add_(1)
        g["len"][:Bp].add_(1)
        g["blk_i"].add_(1)
        nxt = torch.remainder(g["blk_i"], self.BLOCK_MAX)
        g["slot"][:Bp].copy_(g["slot_sched"].index_select(0, nxt)[0, :Bp])
        return tokens, lps

    def _get_graph(self, Bp: int, temp: float):
        key = (Bp, temp)
        if key in self._graphs:
            return self._graphs[key]
        # tune the hipblaslt algo for THIS padded batch's GEMM shapes before
        # capture — the default heuristic has a ~20 us/call floor at skinny M
        ops.pretune_decode_shapes(ops.decode_gemm_shapes(self.cfg, [Bp]),
                                  device=self.device)
        if self.fp8_decode:
            # projections only — logits/sampling stay bf16
            ops.pretune_fp8_decode_shapes(
                [s for s in ops.decode_gemm_shapes(self.cfg, [Bp])
                 if s[1] != self.cfg.vocab_size], device=self.device)
        # warm up eager once (cuBLAS-style workspace init), then capture
        torch.cuda.synchronize()
        self._decode_compute(Bp, temp)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        if self._graph_pool is None:
            with torch.cuda.graph(graph):
                tokens, lps = self._decode_compute(Bp, temp)
            self._graph_pool = graph.pool()
        else:
            with torch.cuda.graph(graph, pool=self._graph_pool):
                tokens, lps = self._decode_compute(Bp, temp)
        self._graphs[key] = (graph, tokens, lps)
        return self._graphs[key]

    def _block_len(self, batch: list[Sequence]) -> int:
        """Steps to run without a host check: bounded by max_tokens /
        context headroom per seq, and by EOS_BLOCK when stop tokens are
        live (over-decode past a stop is computed but never appended)."""
        R = self.BLOCK_MAX
        if self.eos_token_id is not None or any(s.params.stop_token_ids for s in batch):
            R = self.EOS_BLOCK
        for s in batch:
            R = min(R,
                    s.params.max_tokens - len(s.output_ids),
                    self.max_model_len - s.total_len)
        R = max(1, R)
        # shrink until the whole block's page reservation fits the pool —
        # a tight pool must behave like single-step decode (preemption
        # logic unchanged), not livelock on an oversized reservation
        free = self._effective_free_pages()
        while R > 1:
            need = sum(max(0, KVCache.pages_needed(s.total_len + R) - len(s.pages))
                       for s in batch)
            if need <= free:
                break
            R //= 2
        return R

    def _run_decode_graph(self, batch: list[Sequence], temp: float) -> int:
        if self._gb is None:
            self._init_graph_buffers()
        B = len(batch)
        Bp = next(s for s in self._GRAPH_SIZES if s >= B)
        hb, gb = self._hb, self._gb
        R = self._block_len(batch)

        tok_np = hb["tok"].numpy()
        pos_np = hb["pos"].numpy()
        sched_np = hb["slot_sched"].numpy()
        len_np = hb["len"].numpy()
        bt_np = hb["bt"].numpy()
        for i, seq in enumerate(batch):
            if seq.state != SeqState.RUNNING:
                return 0  # preempted mid-loop; retry next step with new batch
            pos = seq.total_len
            # reserve pages for the WHOLE block up front (no host work mid-block)
            if not self._grow_pages(seq, pos + R - 1):
                return 0  # batch composition changed; retry next step
            tok_np[i] = seq.last_token
            pos_np[i] = pos
            for r in range(R):
                p = pos + r
                sched_np[r, i] = seq.pages[p // PAGE_SIZE] * PAGE_SIZE + p % PAGE_SIZE
            len_np[i] = pos + 1
            np_pages = seq.pages
            bt_np[i, : len(np_pages)] = np_pages
        # pad rows: token 0, pos 0, slot 0 (reserved null page), len 1, bt row 0
        if Bp > B:
            tok_np[B:Bp] = 0
            pos_np[B:Bp] = 0
            sched_np[:, B:Bp] = 0
            len_np[B:Bp] = 1
            bt_np[B:Bp, 0] = 0

        # a LATER seq's page grab may have preempted an EARLIER, already-
        # staged seq — its staged slots would point at freed pages. Bail
        # and retry with the new batch composition.
        if any(s.state != SeqState.RUNNING for s in batch):
            return 0

        def stage():
            for k in ("tok", "pos", "len"):
                gb[k][:Bp].copy_(hb[k][:Bp], non_blocking=True)
            gb["bt"][:Bp].copy_(hb["bt"][:Bp], non_blocking=True)
            gb["slot_sched"][:, :Bp].copy_(hb["slot_sched"][:, :Bp], non_blocking=True)
            gb["blk_i"].zero_()
            gb["slot"][:Bp].copy_(gb["slot_sched"][0, :Bp])

        stage()
        fresh = (Bp, temp) not in self._graphs
        graph, _tokens_out, _lps_out = self._get_graph(Bp, temp)
        if fresh:
            # capture ran the compute (incl. state advance) on live buffers —
            # restage so replay 0 starts from this block's real inputs
            stage()
        for _ in range(R):
            graph.replay()

        tok_hist = gb["tok_hist"][:R, :B].tolist()
        lp_hist = gb["lp_hist"][:R, :B].tolist()
        self._sample_step += R
        n_tokens = 0
        for i, seq in enumerate(batch):
            for r in range(R):
                if seq.state != SeqState.RUNNING:
                    break  # finished mid-block: later tokens computed but dropped
                seq.output_ids.append(int(tok_hist[r][i]))
                seq.logprobs.append(float(lp_hist[r][i]))
                n_tokens += 1
                self._maybe_finish(seq)
        return n_tokens

    # ------------------------------------------------------------------
    def _sample_and_append(self, batch: list[Sequence], hidden: torch.Tensor) -> None:
        logits = self.model.logits(hidden)  # [B, V] bf16
        # group rows by (temperature, top_k, top_p); default path is one fused call
        temp = batch[0].params.temperature
        uniform = all(s.params.temperature == temp and s.params.top_k <= 0 and s.params.top_p >= 1.0
                      for s in batch)
        self._sample_step += 1
        if uniform:
            tokens, logprobs = ops.sample_logprob(logits.contiguous(), temp, self.seed, self._sample_step)
            tokens = tokens.tolist()
            logprobs = logprobs.tolist()
        else:
            tokens, logprobs = [], []
            for i, seq in enumerate(batch):
                row = logits[i : i + 1]
                p = seq.params
                if p.top_k > 0 or p.top_p < 1.0:
                    row = _mask_top_k_top_p(row.float(), p.top_k, p.top_p, p.temperature).to(torch.bfloat16)
                t, lp = ops.sample_logprob(row.contiguous(), p.temperature, self.seed + i, self._sample_step)
                tokens.append(int(t[0]))
                logprobs.append(float(lp[0]))

        for seq, tok, lp in zip(batch, tokens, logprobs):
            seq.output_ids.append(int(tok))
            seq.logprobs.append(float(lp))
            self._maybe_finish(seq)

    def _maybe_finish(self, seq: Sequence) -> None:
        tok = seq.output_ids[-1]
        stop_ids = set(seq.params.stop_token_ids)
        if self.eos_token_id is not None:
            stop_ids.add(self.eos_token_id)
        if tok in stop_ids:
            self._finish(seq, "stop")
        elif len(seq.output_ids) >= seq.params.max_tokens:
            self._finish(seq, "length")
        elif seq.total_len >= self.max_model_len:
            # context-window cap: stop before the block table would overflow
            self._finish(seq, "length")

    def _finish(self, seq: Sequence, reason: str) -> None:
        seq.state = SeqState.FINISHED
        seq.finish_reason = reason
        if seq.pages:
            if (self.prefix_cache is not None and reason in ("stop", "length")
                    and seq.weight_version == self._weight_version):
                # make this sequence's full pages (prompt + generated)
                # available to future requests — multi-turn rollouts extend
                # exactly this token prefix. Skipped when the weights moved
                # mid-generation (async training): that KV is stale.
                self.prefix_cache.publish(seq.prompt_ids + seq.output_ids,
                                          seq.pages, seq.n_cached_pages)
            self._release_pages(seq.pages)
            seq.pages = []
        self.finished[seq.request_id] = seq

    # ------------------------------------------------------------------
    def generate(self, prompts: list[list[int]], params: SamplingParams | list[SamplingParams],
                 on_step: Callable[[], None] | None = None) -> list[RequestOutput]:
        """Blocking batch generation (bench / tests)."""
        if isinstance(params, SamplingParams):
            params = [params] * len(prompts)
        for i, (p, sp) in enumerate(zip(prompts, params)):
            self.add_request(f"gen-{i}", p, sp)
        while self.has_unfinished():
            self.step()
            if on_step is not None:
                on_step()
        outs = {o.request_id: o for o in self.pop_finished()}
        return [outs[f"gen-{i}"] for i in range(len(prompts))]


def _mask_top_k_top_p(logits_f32: torch.Tensor, top_k: int, top_p: float, temperature: float) -> torch.Tensor:
    """Pre-mask logits (-inf) for top-k / nucleus sampling; the fused kernel
    then samples the masked distribution exactly. Rare path — the training
    defaults are temp 1.0 / top_p 1.0 (reference base.yaml:49-60)."""
    z = logits_f32 / max(temperature, 1e-6)
    if top_k > 0 and top_k < z.shape[-1]:
        kth = torch.topk(z, top_k, dim=-1).values[..., -1:]
        logits_f32 = logits_f32.masked_fill(z < kth, float("-inf"))
        z = z.masked_fill(z < kth, float("-inf"))
    if top_p < 1.0:
        sorted_z, idx = torch.sort(z, descending=True, dim=-1)
        probs = torch.softmax(sorted_z, dim=-1)
        cum = probs.cumsum(-1)
        keep_sorted = cum - probs < top_p  # always keep the top token
        keep = torch.zeros_like(keep_sorted).scatter(-1, idx, keep_sorted)
        logits_f32 = logits_f32.masked_fill(~keep, float("-inf"))
    return logits_f32
