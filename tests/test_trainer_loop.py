"""Full training-loop integration on CPU (BASELINE config 1 shape):
UnifiedTrainer + CPUBackend + real gateway + @rollout/@evaluator.

Covers: the 8-stage batch, GRPO advantages landing in the update,
metrics, checkpoint save/resume, validation, rejection sampling skip.
"""

import sys
from pathlib import Path

import httpx
import pytest

sys.path.insert(0, str(Path(__file__).parent))

import rllm_amd
from rllm_amd.data.dataset import Dataset
from rllm_amd.trainer.algorithms.config import AlgorithmConfig
from rllm_amd.trainer.cpu_backend import CPUBackend
from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer


@rllm_amd.rollout
def solve_flow(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user", "content": str(task.instruction)}]},
                   timeout=30.0)
    r.raise_for_status()
    return None


@rllm_amd.evaluator
def parity_eval(task, episode):
    """Reward: fraction of generated token ids that are even — a dense
    deterministic signal with within-group variance (sampling temp 1.0)."""
    step = episode.trajectories[0].steps[-1]
    if not step.response_ids:
        return 0.0
    frac = sum(1 for t in step.response_ids if t % 2 == 0) / len(step.response_ids)
    return float(frac)


def make_dataset(n=4):
    return Dataset([{"question": f"task {i}", "id": str(i)} for i in range(n)]).as_tasks(id_key="id")


def test_trainer_full_loop(tmp_path):
    backend = CPUBackend(solve_flow, parity_eval, rollout_max_tokens=8, seed=0)
    cfg = TrainerConfig(
        total_epochs=1, train_batch_size=2, rollout_n=4, max_steps=2,
        checkpoint_dir=str(tmp_path / "ckpt"), save_freq=2,
        episode_log_dir=str(tmp_path / "episodes"), logger_backends=["jsonl"],
    )
    trainer = UnifiedTrainer(backend, make_dataset(4), config=cfg,
                             algorithm_config=AlgorithmConfig())
    trainer.fit()

    assert trainer.state.global_step == 2
    assert trainer.state.total_episodes == 2 * 2 * 4  # steps * tasks * n
    # checkpoint written
    assert (tmp_path / "ckpt" / "global_step_2" / "actor.pt").exists()
    assert (tmp_path / "ckpt" / "latest_checkpointed_iteration.txt").read_text() == "2"
    # episode logs in the canonical format
    ep_dirs = list((tmp_path / "episodes").glob("train_step_*"))
    assert len(ep_dirs) == 2
    from rllm_amd.utils.episode_logger import EpisodeLogger

    eps = EpisodeLogger.load_episodes(ep_dirs[0])
    assert len(eps) == 8
    assert all(len(e.trajectories[0].steps[0].response_ids) > 0 for e in eps)
    # advantages were written onto steps (GRPO ran)
    advs = [e.trajectories[0].steps[0].advantage for e in eps]
    assert any(a not in (None, 0.0) for a in advs)


def test_trainer_resume(tmp_path):
    ds = make_dataset(4)
    cfg = TrainerConfig(total_epochs=2, train_batch_size=2, rollout_n=2, max_steps=2,
                        checkpoint_dir=str(tmp_path / "ck"), save_freq=1,
                        logger_backends=[])
    b1 = CPUBackend(solve_flow, parity_eval, rollout_max_tokens=6, seed=0)
    t1 = UnifiedTrainer(b1, ds, config=cfg)
    t1.fit()
    assert t1.state.global_step == 2

    cfg2 = TrainerConfig(total_epochs=2, train_batch_size=2, rollout_n=2, max_steps=3,
                         checkpoint_dir=str(tmp_path / "ck"), save_freq=10,
                         resume="auto", logger_backends=[])
    b2 = CPUBackend(solve_flow, parity_eval, rollout_max_tokens=6, seed=0)
    t2 = UnifiedTrainer(b2, ds, config=cfg2)
    t2.fit()
    assert t2.state.global_step == 3  # continued from 2


def test_trainer_validation(tmp_path):
    backend = CPUBackend(solve_flow, parity_eval, rollout_max_tokens=6, seed=1)
    cfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=2, max_steps=1,
                        val_freq=1, rollout_n_val=1, logger_backends=[])
    trainer = UnifiedTrainer(backend, make_dataset(2), val_dataset=make_dataset(2), config=cfg)
    trainer.fit()
    # validation ran without blowing up; metrics tracked on state
    assert trainer.state.global_step == 1


def test_cpu_backend_cumulative_multiturn(tmp_path):
    """Config-1 shape WITH cumulative token mode: whole pipeline on CPU,
    two-turn flow, merged packed rows (prefix-extension end-to-end with no
    GPU in the loop)."""
    import httpx

    import rllm_amd
    from rllm_amd.data.dataset import Dataset
    from rllm_amd.gateway.models import GatewayConfig
    from rllm_amd.trainer.cpu_backend import CPUBackend
    from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer

    @rllm_amd.rollout
    def two_turn(task, config):
        msgs = [{"role": "user", "content": str(task.instruction)}]
        r1 = httpx.post(config.base_url + "/chat/completions",
                        json={"model": config.model, "messages": msgs, "max_tokens": 4},
                        timeout=60.0)
        r1.raise_for_status()
        reply = r1.json()["choices"][0]["message"]["content"]
        msgs += [{"role": "assistant", "content": reply},
                 {"role": "user", "content": "and?"}]
        httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model, "messages": msgs, "max_tokens": 4},
                   timeout=60.0).raise_for_status()

    @rllm_amd.evaluator
    def ev(task, episode):
        return 1.0 if episode.trajectories and episode.trajectories[0].steps else 0.0

    backend = CPUBackend(two_turn, ev, n_parallel_tasks=4, rollout_max_tokens=4,
                         gateway_config=GatewayConfig(cumulative_token_mode=True))
    tasks = Dataset([{"question": f"q{i}", "id": str(i)} for i in range(2)]).as_tasks(id_key="id")
    tcfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=2, max_steps=1,
                         logger_backends=[])
    trainer = UnifiedTrainer(backend, tasks, config=tcfg)
    trainer.fit()
    assert trainer.state.global_step == 1


def test_context_curriculum_raises_rollout_cap(tmp_path):
    """DeepScaleR-style context schedule: the trainer raises the backend's
    rollout response cap at the configured global steps."""
    backend = CPUBackend(solve_flow, parity_eval, rollout_max_tokens=4, seed=0)
    cfg = TrainerConfig(total_epochs=4, train_batch_size=2, rollout_n=2, max_steps=3,
                        logger_backends=[],
                        context_curriculum=[[0, 6], [2, 10]])
    trainer = UnifiedTrainer(backend, make_dataset(2), config=cfg)
    seen = []
    orig = backend.set_max_response_tokens

    def spy(n):
        seen.append((trainer.state.global_step, n))
        orig(n)

    backend.set_max_response_tokens = spy
    trainer.fit()
    assert seen == [(0, 6), (2, 10)]
    assert backend.rollout_max_tokens == 10
    assert backend.flow_engine.default_sampling_params["max_tokens"] == 10
