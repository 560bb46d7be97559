"""GPU: LoRA training on the kernel substrate — adapter-only GRPO update
(flat optimizer over adapters), ref-from-base KL (no second model copy),
and merged-adapter decode consistency with the train path."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")

from rllm_amd.models import lora  # noqa: E402
from rllm_amd.models.config import ModelConfig  # noqa: E402

CFG = ModelConfig(name="lora-tiny", hidden_size=512, intermediate_size=1024,
                  num_layers=2, num_heads=8, num_kv_heads=2, head_dim=128,
                  vocab_size=1024, tie_word_embeddings=False)


def _rows(n=6, seed=0):
    import random

    from rllm_amd.trainer.batch import PackedRow

    rng = random.Random(seed)
    rows = []
    for _ in range(n):
        L = rng.randint(12, 24)
        p = rng.randint(4, 8)
        toks = [rng.randrange(CFG.vocab_size) for _ in range(L)]
        mask = [0] * p + [1] * (L - p)
        rows.append(PackedRow(tokens=toks, response_mask=mask,
                              advantages=[rng.uniform(-1, 1)] * L,
                              rollout_logprobs=[-1.0] * L))
    return rows


@requires_gpu
def test_lora_grpo_update_trains_only_adapters():
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.trainer.policy import PolicyTrainer, PolicyTrainerConfig

    model = QwenModel(CFG, device="cuda").init_random(seed=1)
    base_qkv = model.layers[0].qkv_proj.detach().clone()
    lora.inject_lora(model, lora.LoRAConfig(r=8, alpha=16))
    pt = PolicyTrainer(model, ref_model=None,
                       config=PolicyTrainerConfig(lr=1e-3, kl_beta=1e-2, use_ref=True,
                                                  ref_from_lora_base=True,
                                                  old_logprob_mode="alias"))
    # optimizer state covers adapters only
    n_adapters = sum(p.numel() for p in lora.lora_parameters(model))
    assert pt.flat_param.numel() == n_adapters

    rows = _rows()
    m1 = pt.update_policy(rows)
    m2 = pt.update_policy(rows)
    assert m1["actor/grad_norm"] > 0
    assert m2["actor/grad_norm"] > 0
    # base weights untouched; adapters moved
    assert torch.equal(model.layers[0].qkv_proj, base_qkv)
    moved = sum(float(p.detach().abs().sum()) for n, p in model.layers[0].lora.items() if n.endswith("_B"))
    assert moved > 0
    # KL vs base is finite and the ref pass didn't flip adapter state
    assert model.lora_enabled


@requires_gpu
def test_lora_merged_decode_matches_train_logits():
    """Merged-adapter rollout path == unmerged train path (bf16 tolerance):
    prefill one short prompt after merge_lora_, compare last-token logits
    against forward_train with adapters active."""
    from rllm_amd.engine.inference.kv_cache import KVCache
    from rllm_amd.models.qwen import QwenModel, make_prefill_tiles

    model = QwenModel(CFG, device="cuda").init_random(seed=2)
    lora.inject_lora(model, lora.LoRAConfig(r=8, alpha=16, seed=5))
    # give the adapters real weight
    gen = torch.Generator().manual_seed(9)
    for layer in model.layers:
        for k, p in layer.lora.items():
            if k.endswith("_B"):
                with torch.no_grad():
                    p.copy_((torch.randn(p.shape, generator=gen) * 0.02).to(p.dtype))

    T = 18
    ids = torch.randint(0, CFG.vocab_size, (T,), device="cuda")
    pos = torch.arange(T, device="cuda", dtype=torch.int32)
    cu = [0, T]

    hidden_train = model.forward_train(ids, pos, cu)
    logits_train = model.logits(hidden_train[-1:]).float()

    lora.merge_lora_(model)
    cache = KVCache(CFG.num_layers, CFG.num_kv_heads, CFG.head_dim,
                    num_pages=32, device="cuda")
    pages = cache.alloc((T + 15) // 16)
    slot_mapping = torch.tensor(
        [pages[i // 16] * 16 + i % 16 for i in range(T)], device="cuda", dtype=torch.int32)
    tiles = make_prefill_tiles([T], "cuda")
    hidden_prefill = model.forward_prefill(ids, pos, tiles, cache, slot_mapping)
    logits_prefill = model.logits(hidden_prefill[-1:]).float()

    assert torch.allclose(logits_train, logits_prefill, atol=0.1, rtol=0.05), \
        (logits_train - logits_prefill).abs().max()

    # unmerge restores the base path
    lora.unmerge_lora_(model)
    with lora.disabled(model):
        hidden_base = model.forward_train(ids, pos, cu)
    assert not torch.allclose(model.logits(hidden_base[-1:]).float(), logits_train,
                              atol=1e-3, rtol=1e-3)


@requires_gpu
def test_native_backend_lora_full_loop(tmp_path):
    """Whole framework loop with LoRA: rollouts through the shared-weight
    engine with adapters active, adapter-only GRPO update, KL vs the
    adapter-disabled base."""
    import sys
    from pathlib import Path as _P

    sys.path.insert(0, str(_P(__file__).parent))
    import httpx

    import rllm_amd
    from rllm_amd.data.dataset import Dataset
    from rllm_amd.trainer.native_backend import NativeBackend
    from rllm_amd.trainer.policy import PolicyTrainerConfig
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer

    @rllm_amd.rollout
    def lora_flow(task, config):
        r = httpx.post(config.base_url + "/chat/completions",
                       json={"model": config.model,
                             "messages": [{"role": "user", "content": str(task.instruction)}],
                             "max_tokens": 12},
                       timeout=120.0)
        r.raise_for_status()
        return None

    @rllm_amd.evaluator
    def lora_eval(task, episode):
        step = episode.trajectories[0].steps[-1]
        return 1.0 if step.response_ids and step.response_ids[0] % 2 == 0 else 0.0

    backend = NativeBackend(
        lora_flow, lora_eval, model_config=CFG,
        policy_config=PolicyTrainerConfig(lr=1e-3, kl_beta=1e-3),
        kv_budget_bytes=64 << 20, lora=lora.LoRAConfig(r=8, alpha=16),
        rollout_sampling_params={"temperature": 1.0, "max_tokens": 12},
        n_parallel_tasks=8, seed=4)
    tasks = Dataset([{"question": f"q {i}", "id": str(i)} for i in range(4)]).as_tasks(id_key="id")
    tcfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=4, max_steps=2,
                         checkpoint_dir=str(tmp_path / "ck"), save_freq=0, logger_backends=[])
    trainer = UnifiedTrainer(backend, tasks, config=tcfg)
    trainer.fit()

    assert trainer.state.global_step == 2
    assert backend.ref_model is None                 # no second copy
    assert backend.policy.cfg.ref_from_lora_base
    moved = sum(float(p.detach().abs().sum())
                for layer in backend.model.layers
                for n, p in layer.lora.items() if n.endswith("_B"))
    assert moved > 0                                  # adapters trained
