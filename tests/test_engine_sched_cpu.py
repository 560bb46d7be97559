"""CPU tests for the engine's HOST-side scheduling logic (no kernels):
prefill admission, block-length selection for block decode, and page
accounting — the logic the GPU integration tests exercise end to end."""

import pytest
import torch

from rllm_amd.engine.inference.kv_cache import PAGE_SIZE, KVCache
from rllm_amd.engine.inference.llm_engine import LLMEngine, SamplingParams, SeqState
from rllm_amd.models.config import ModelConfig
from rllm_amd.models.qwen import QwenModel


def make_engine(pages=64, eos=None, max_model_len=512, batched_tokens=128):
    cfg = ModelConfig(name="sched-tiny", hidden_size=64, intermediate_size=128,
                      num_layers=1, num_heads=4, num_kv_heads=2, head_dim=16,
                      vocab_size=128)
    m = QwenModel(cfg, device="cpu").init_random(seed=0)
    kv = KVCache(cfg.num_layers, cfg.num_kv_heads, cfg.head_dim, pages, device="cpu")
    return LLMEngine(m, kv_cache=kv, eos_token_id=eos, use_hip_graph=False,
                     max_model_len=max_model_len, max_num_batched_tokens=batched_tokens,
                     enable_prefix_caching=False)


def _admit(e, n_seqs=2, prompt=40, max_tokens=100):
    for i in range(n_seqs):
        e.add_request(f"s{i}", list(range(2, 2 + prompt)), SamplingParams(max_tokens=max_tokens))
    batch = e._schedule_prefill()
    for s in batch:
        s.state = SeqState.RUNNING
        e.running.append(s)
    return batch


def test_prefill_admission_token_budget():
    e = make_engine(batched_tokens=100)
    for i in range(4):
        e.add_request(f"s{i}", list(range(2, 62)), SamplingParams(max_tokens=8))
    b1 = e._schedule_prefill()
    assert len(b1) == 1  # 60 tokens; a second 60 would exceed 100
    b2 = e._schedule_prefill()
    assert len(b2) == 1
    # pages were allocated for prompt+1
    assert len(b1[0].pages) == KVCache.pages_needed(61)


def test_prefill_rejects_impossible_prompt():
    e = make_engine(pages=4)  # 4 pages = 64 tokens max
    e.add_request("big", list(range(2, 120)), SamplingParams(max_tokens=8))
    assert e._schedule_prefill() == []
    out = e.pop_finished()
    assert out and out[0].finish_reason == "error"


def test_block_len_caps():
    e = make_engine(pages=64, eos=None)
    batch = _admit(e, n_seqs=1, prompt=40, max_tokens=100)
    # no stop tokens: full block
    assert e._block_len(batch) == e.BLOCK_MAX
    # stop tokens live: EOS_BLOCK
    e2 = make_engine(pages=64, eos=5)
    b2 = _admit(e2, n_seqs=1)
    assert e2._block_len(b2) == e2.EOS_BLOCK
    # near max_tokens: capped by remaining budget
    batch[0].params.max_tokens = len(batch[0].output_ids) + 3
    assert e._block_len(batch) == 3
    # near the context window: capped by model_len headroom
    batch[0].params.max_tokens = 10_000
    e.max_model_len = batch[0].total_len + 2
    assert e._block_len(batch) == 2


def test_block_len_shrinks_to_fit_page_pool():
    e = make_engine(pages=8, eos=None)  # tiny pool
    batch = _admit(e, n_seqs=2, prompt=30, max_tokens=500)
    free = e._effective_free_pages()
    R = e._block_len(batch)
    need = sum(max(0, KVCache.pages_needed(s.total_len + R) - len(s.pages)) for s in batch)
    assert need <= free
    assert R >= 1
    # a roomy pool would not shrink
    e_big = make_engine(pages=256, eos=None)
    b_big = _admit(e_big, n_seqs=2, prompt=30, max_tokens=500)
    assert e_big._block_len(b_big) == e_big.BLOCK_MAX


def test_grow_pages_reserves_block():
    e = make_engine(pages=64)
    batch = _admit(e, n_seqs=1, prompt=30, max_tokens=400)
    seq = batch[0]
    before = len(seq.pages)
    assert e._grow_pages(seq, seq.total_len + 15)
    assert len(seq.pages) == KVCache.pages_needed(seq.total_len + 16)
    assert len(seq.pages) > before


def test_finish_releases_pages_and_records_reason():
    e = make_engine(pages=64, eos=7)
    batch = _admit(e, n_seqs=1, prompt=20, max_tokens=4)
    seq = batch[0]
    free0 = e.kv.num_free_pages
    seq.output_ids.extend([1, 2, 3, 4])
    e._maybe_finish(seq)
    assert seq.state == SeqState.FINISHED and seq.finish_reason == "length"
    assert e.kv.num_free_pages > free0
