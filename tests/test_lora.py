"""LoRA adapters (models/lora.py): injection, forward math, merge/unmerge
round-trip, state dict, disabled() reference mode — CPU (the _lin helper is
plain torch; kernels are only on the normalization/attention path)."""

import pytest
import torch

from rllm_amd.models import lora
from rllm_amd.models.config import MODEL_CONFIGS
from rllm_amd.models.qwen import QwenModel, _linear


@pytest.fixture()
def model():
    m = QwenModel(MODEL_CONFIGS["tiny"], device="cpu").init_random(seed=3)
    return m


def _activate(m):
    """Nonzero B so the adapter actually changes outputs."""
    gen = torch.Generator().manual_seed(7)
    for layer in m.layers:
        for k, p in layer.lora.items():
            if k.endswith("_B"):
                with torch.no_grad():
                    p.copy_(torch.randn(p.shape, generator=gen).to(p.dtype) * 0.05)


def test_inject_freezes_base_and_counts(model):
    n_total = sum(p.numel() for p in model.parameters())
    cfg = lora.inject_lora(model, lora.LoRAConfig(r=4, alpha=8))
    trainable = lora.lora_parameters(model)
    assert all(p.requires_grad for p in trainable)
    n_train = sum(p.numel() for p in trainable)
    expect = 0
    for layer in model.layers:
        for name in cfg.targets:
            out_f, in_f = getattr(layer, name).shape
            expect += cfg.r * (in_f + out_f)
    assert n_train == expect
    assert n_train < n_total
    frozen = [p for p in model.parameters() if not p.requires_grad]
    assert sum(p.numel() for p in frozen) == n_total


def test_zero_init_is_identity(model):
    x = torch.randn(5, model.cfg.hidden_size).to(torch.bfloat16)
    layer = model.layers[0]
    base = _linear(x, layer.qkv_proj, layer.qkv_bias)
    lora.inject_lora(model, lora.LoRAConfig(r=4))
    model._stamp_lora()
    assert torch.equal(layer._lin(x, "qkv_proj", layer.qkv_bias), base)


def test_forward_math_and_disabled(model):
    cfg = lora.inject_lora(model, lora.LoRAConfig(r=4, alpha=8))
    _activate(model)
    model._stamp_lora()
    layer = model.layers[1]
    x = torch.randn(3, model.cfg.q_size).to(torch.bfloat16)
    base = _linear(x, layer.o_proj)
    got = layer._lin(x, "o_proj")
    expect = base + (x @ layer.lora["o_proj_A"].t()) @ layer.lora["o_proj_B"].t() * cfg.scale
    assert torch.allclose(got.float(), expect.float())
    assert not torch.equal(got, base)
    with lora.disabled(model):
        model._stamp_lora()
        assert torch.equal(layer._lin(x, "o_proj"), base)
    model._stamp_lora()
    assert torch.equal(layer._lin(x, "o_proj"), got)


def test_merge_unmerge_roundtrip(model):
    lora.inject_lora(model, lora.LoRAConfig(r=4, alpha=8))
    _activate(model)
    model._stamp_lora()
    layer = model.layers[0]
    x = torch.randn(4, model.cfg.hidden_size).to(torch.bfloat16)
    unmerged = layer._lin(x, "qkv_proj", layer.qkv_bias)
    w_before = layer.qkv_proj.detach().clone()
    lora.merge_lora_(model)
    model._stamp_lora()
    # merged: the plain GEMM now carries the adapter
    merged = layer._lin(x, "qkv_proj", layer.qkv_bias)
    assert torch.allclose(merged.float(), unmerged.float(), atol=0.05, rtol=0.05)
    assert not torch.equal(layer.qkv_proj, w_before)
    # merge is idempotent
    lora.merge_lora_(model)
    lora.unmerge_lora_(model)
    # fp32-accumulated round trip restores base weights to bf16 exactness
    assert torch.allclose(layer.qkv_proj.float(), w_before.float(), atol=1e-2)
    lora.unmerge_lora_(model)  # idempotent
    assert torch.allclose(layer.qkv_proj.float(), w_before.float(), atol=1e-2)


def test_disabled_while_merged_raises(model):
    lora.inject_lora(model)
    lora.merge_lora_(model)
    with pytest.raises(RuntimeError):
        with lora.disabled(model):
            pass


def test_state_dict_roundtrip(model):
    lora.inject_lora(model, lora.LoRAConfig(r=4))
    _activate(model)
    sd = lora.lora_state_dict(model)
    assert len(sd) == len(model.layers) * len(lora.TARGETS_ALL) * 2
    m2 = QwenModel(MODEL_CONFIGS["tiny"], device="cpu").init_random(seed=3)
    lora.inject_lora(m2, lora.LoRAConfig(r=4))
    lora.load_lora_state_dict(m2, sd)
    for la, lb in zip(model.layers, m2.layers):
        for k in la.lora:
            assert torch.equal(la.lora[k], lb.lora[k])


def test_flatten_params_sees_only_adapters(model):
    from rllm_amd.trainer.optim import flatten_params

    cfg = lora.inject_lora(model, lora.LoRAConfig(r=4))
    flat, flat_grad = flatten_params(model)
    n_adapters = sum(p.numel() for p in lora.lora_parameters(model))
    assert flat.numel() == n_adapters
    # grads land in the flat buffer
    layer = model.layers[0]
    x = torch.randn(2, model.cfg.q_size).to(torch.bfloat16)
    model._stamp_lora()
    y = layer._lin(x, "o_proj")
    y.float().sum().backward()
    assert flat_grad.abs().sum() > 0
    assert layer.o_proj.grad is None or layer.o_proj.grad.abs().sum() == 0
