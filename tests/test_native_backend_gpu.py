"""GPU integration: UnifiedTrainer + NativeBackend end-to-end — the full
framework loop (gateway → flow → native engine → traces → GRPO update)
on the MI355X substrate with a small Qwen-architecture model."""

import sys
from pathlib import Path

import httpx
import pytest
import torch

sys.path.insert(0, str(Path(__file__).parent))

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")

import rllm_amd  # noqa: E402
from rllm_amd.data.dataset import Dataset  # noqa: E402


@rllm_amd.rollout
def gpu_flow(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user", "content": str(task.instruction)}],
                         "max_tokens": 16},
                   timeout=120.0)
    r.raise_for_status()
    return None


@rllm_amd.evaluator
def gpu_eval(task, episode):
    step = episode.trajectories[0].steps[-1]
    if not step.response_ids:
        return 0.0
    return float(sum(1 for t in step.response_ids if t % 2 == 0) / len(step.response_ids))


@requires_gpu
def test_native_backend_full_loop(tmp_path):
    from rllm_amd.models.config import ModelConfig
    from rllm_amd.trainer.native_backend import NativeBackend
    from rllm_amd.trainer.policy import PolicyTrainerConfig
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer

    cfg = ModelConfig(name="gpu-loop-tiny", hidden_size=512, intermediate_size=1024,
                      num_layers=2, num_heads=8, num_kv_heads=2, head_dim=128,
                      vocab_size=1024, tie_word_embeddings=False)
    backend = NativeBackend(
        gpu_flow, gpu_eval, model_config=cfg,
        policy_config=PolicyTrainerConfig(lr=1e-4, kl_beta=1e-3, grad_clip=1.0),
        kv_budget_bytes=64 << 20,
        rollout_sampling_params={"temperature": 1.0, "max_tokens": 16},
        n_parallel_tasks=8, seed=3)

    tasks = Dataset([{"question": f"task {i}", "id": str(i)} for i in range(4)]).as_tasks(id_key="id")
    tcfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=4, max_steps=2,
                         checkpoint_dir=str(tmp_path / "ck"), save_freq=2,
                         logger_backends=[])
    trainer = UnifiedTrainer(backend, tasks, config=tcfg)
    trainer.fit()

    assert trainer.state.global_step == 2
    assert trainer.state.weight_version == 2
    # engine saw the weight-version bump (colocated sync)
    assert backend.engine.weight_version == 2
    # checkpoint with optimizer state exists
    assert (tmp_path / "ck" / "global_step_2" / "actor.pt").exists()
    # KV pages all reclaimed after training (evict idle cached pages first)
    if backend.engine.prefix_cache is not None:
        backend.engine.prefix_cache.evict(1 << 30)
    assert backend.engine.kv.num_free_pages == backend.engine.kv.num_pages - 1


@requires_gpu
def test_sft_trainer_reduces_nll():
    from rllm_amd.models.config import ModelConfig
    from rllm_amd.models.qwen import QwenModel
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.trainer.sft import SFTConfig, SFTTrainer, rows_from_messages
    from rllm_amd.utils.tokenizer import ByteTokenizer

    cfg = ModelConfig(name="sft-tiny", hidden_size=512, intermediate_size=1024,
                      num_layers=2, num_heads=8, num_kv_heads=2, head_dim=128,
                      vocab_size=1024, tie_word_embeddings=False)
    model = QwenModel(cfg, device="cuda").init_random(seed=5)
    parser = QwenChatTemplateParser(ByteTokenizer())
    msgs = [[{"role": "user", "content": f"q{i}"}, {"role": "assistant", "content": "the answer is 42"}]
            for i in range(8)]
    rows = rows_from_messages(msgs, parser)
    trainer = SFTTrainer(model, SFTConfig(lr=1e-3, epochs=1))
    m1 = trainer.train_step(rows)
    for _ in range(4):
        m2 = trainer.train_step(rows)
    assert m2["sft/nll"] < m1["sft/nll"], (m1, m2)


@rllm_amd.rollout
def two_turn_flow(task, config):
    """Multi-turn agent: two chat calls with accumulated history (the
    FrozenLake/tool-agent shape, BASELINE config 4)."""
    msgs = [{"role": "user", "content": f"move for {task.instruction}?"}]
    r1 = httpx.post(config.base_url + "/chat/completions",
                    json={"model": config.model, "messages": msgs, "max_tokens": 8},
                    timeout=120.0)
    r1.raise_for_status()
    reply = r1.json()["choices"][0]["message"]["content"]
    msgs = msgs + [{"role": "assistant", "content": reply},
                   {"role": "user", "content": "grid changed. next move?"}]
    r2 = httpx.post(config.base_url + "/chat/completions",
                    json={"model": config.model, "messages": msgs, "max_tokens": 8},
                    timeout=120.0)
    r2.raise_for_status()
    return None


@requires_gpu
def test_multiturn_cumulative_rloo_native(tmp_path):
    """Config 4 shape: multi-turn flow -> gateway (cumulative token mode) ->
    native engine; RLOO advantages; prefix-merged packed rows in the update."""
    from rllm_amd.gateway.models import GatewayConfig
    from rllm_amd.models.config import ModelConfig
    from rllm_amd.trainer.algorithms.config import AlgorithmConfig
    from rllm_amd.trainer.native_backend import NativeBackend
    from rllm_amd.trainer.policy import PolicyTrainerConfig
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer

    cfg = ModelConfig(name="mt-tiny", hidden_size=512, intermediate_size=1024,
                      num_layers=2, num_heads=8, num_kv_heads=2, head_dim=128,
                      vocab_size=1024, tie_word_embeddings=False)
    backend = NativeBackend(
        two_turn_flow, gpu_eval, model_config=cfg,
        policy_config=PolicyTrainerConfig(lr=1e-4, kl_beta=0.0),
        kv_budget_bytes=64 << 20,
        rollout_sampling_params={"temperature": 1.0, "max_tokens": 8},
        n_parallel_tasks=4, seed=11,
        gateway_config=GatewayConfig(cumulative_token_mode=True))

    from rllm_amd.data.dataset import Dataset

    tasks = Dataset([{"question": f"t{i}", "id": str(i)} for i in range(2)]).as_tasks(id_key="id")
    tcfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=4, max_steps=1,
                         logger_backends=[], episode_log_dir=str(tmp_path / "eps"))
    trainer = UnifiedTrainer(backend, tasks, config=tcfg,
                             algorithm_config=AlgorithmConfig(estimator="rloo"))

    # intercept the backend batch to verify prefix merging
    seen_rows = []
    orig = backend.transform_to_backend_batch

    def capture(groups):
        rows = orig(groups)
        seen_rows.extend(rows)
        return rows

    backend.transform_to_backend_batch = capture
    trainer.fit()

    assert trainer.state.global_step == 1
    assert seen_rows, "no training rows produced"
    # every episode had 2 turns; cumulative mode must merge them into ONE
    # packed row with two action segments (prefix-extension held)
    merged = [r for r in seen_rows if sum(r.response_mask) >= 16]
    assert merged, f"no merged two-turn rows (rows={[(len(r.tokens), sum(r.response_mask)) for r in seen_rows]})"
    row = merged[0]
    # mask pattern: 0s (prompt) 1s (turn1) 0s (obs) 1s (turn2)
    import itertools

    segs = [k for k, _ in itertools.groupby(row.response_mask)]
    assert segs == [0, 1, 0, 1], segs


@requires_gpu
def test_native_backend_from_hf_model_dir(tmp_path):
    """Config-2 real-model path: a local HF dir (config.json + safetensors)
    drives the FULL loop — importer builds the model, rollouts decode, the
    update steps (head_dim 128 so the MFMA kernels engage)."""
    pytest.importorskip("transformers")
    from transformers import Qwen2Config, Qwen2ForCausalLM

    from rllm_amd.trainer.native_backend import NativeBackend
    from rllm_amd.trainer.policy import PolicyTrainerConfig
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer
    from rllm_amd.utils.tokenizer import ByteTokenizer

    torch.manual_seed(3)
    d = tmp_path / "hf"
    Qwen2ForCausalLM(Qwen2Config(
        hidden_size=512, intermediate_size=1024, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, vocab_size=1024,
        max_position_embeddings=4096, tie_word_embeddings=False,
        rope_theta=10000.0)).save_pretrained(str(d), safe_serialization=True)

    backend = NativeBackend(
        gpu_flow, gpu_eval, model_config=str(d),
        tokenizer=ByteTokenizer(),  # the tiny ckpt ships no tokenizer files
        policy_config=PolicyTrainerConfig(lr=1e-4, kl_beta=1e-3, grad_clip=1.0),
        kv_budget_bytes=64 << 20,
        rollout_sampling_params={"temperature": 1.0, "max_tokens": 12},
        n_parallel_tasks=4, seed=5)
    assert backend.hf_model_dir == str(d)
    assert backend.cfg.head_dim == 128 and backend.cfg.num_layers == 2

    tasks = Dataset([{"question": f"t{i}", "id": str(i)} for i in range(2)]).as_tasks(id_key="id")
    trainer = UnifiedTrainer(backend, tasks,
                             config=TrainerConfig(train_batch_size=2, rollout_n=2,
                                                  max_steps=1, logger_backends=[]))
    trainer.fit()
    assert trainer.state.global_step == 1
    # weights really came from the checkpoint (importer parity vs HF tensors)
    import safetensors.torch as st

    sd = st.load_file(str(d / "model.safetensors"))
    got = backend.model.layers[0].down_proj.detach().float().cpu()
    want = sd["model.layers.0.mlp.down_proj.weight"].to(torch.bfloat16).float()
    # the update moved weights by ~lr; they must still be close to the import
    assert (got - want).abs().max() < 0.05
