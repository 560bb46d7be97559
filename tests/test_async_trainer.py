"""Fully-async pipeline tests: coordinator quota/drain, buffer grouping
+ offload, end-to-end FullyAsyncTrainer on the CPU backend."""

import asyncio
import sys
from pathlib import Path

import httpx
import pytest

sys.path.insert(0, str(Path(__file__).parent))

import rllm_amd
from rllm_amd.data.dataset import Dataset
from rllm_amd.trainer.algorithms.config import AsyncTrainingConfig
from rllm_amd.trainer.async_trainer import FullyAsyncTrainer
from rllm_amd.trainer.buffer import TrajectoryGroupBuffer
from rllm_amd.trainer.cpu_backend import CPUBackend
from rllm_amd.trainer.sync_coordinator import SyncCoordinator
from rllm_amd.trainer.unified_trainer import TrainerConfig
from rllm_amd.types import Episode, Step, Trajectory


def make_episode(task_id: str, idx: int, wv: int = 0) -> Episode:
    st = Step(prompt_ids=[1, 2], response_ids=[3, 4], logprobs=[-0.1, -0.2],
              chat_completions=[{"role": "user", "content": "q"}], reward=float(idx % 2),
              weight_version=wv)
    return Episode(id=f"{task_id}:{idx}",
                   trajectories=[Trajectory(name="s", steps=[st], reward=st.reward)])


def test_sync_coordinator_quota_and_drain():
    async def run():
        c = SyncCoordinator(staleness_threshold=0.0, trigger_sync_step=1, mini_batch_size=2)
        assert c.quota == 2
        await c.acquire_dispatch("a")
        await c.acquire_dispatch("b")
        assert not c.can_dispatch()
        # third dispatch blocks until consumption
        acquired = asyncio.Event()

        async def third():
            await c.acquire_dispatch("c")
            acquired.set()

        t = asyncio.create_task(third())
        await asyncio.sleep(0.05)
        assert not acquired.is_set()
        c.mark_consumed(1)
        await asyncio.wait_for(acquired.wait(), 2.0)
        # drain waits for in-flight
        c.mark_done("a")
        c.mark_done("b")
        c.mark_done("c")
        assert await c.drain(timeout=1.0)
        t.cancel()

    asyncio.run(run())


def test_coordinator_pause_blocks_dispatch():
    async def run():
        c = SyncCoordinator(staleness_threshold=10.0)
        c.pause()
        acquired = asyncio.Event()

        async def try_dispatch():
            await c.acquire_dispatch("x")
            acquired.set()

        t = asyncio.create_task(try_dispatch())
        await asyncio.sleep(0.05)
        assert not acquired.is_set()
        c.resume()
        await asyncio.wait_for(acquired.wait(), 2.0)
        t.cancel()

    asyncio.run(run())


def test_buffer_groups_by_task(tmp_path):
    async def run():
        buf = TrajectoryGroupBuffer(group_size=2, offload_dir=str(tmp_path / "off"))
        await buf.add_episode(make_episode("t1", 0, wv=3))
        assert buf.qsize() == 0  # incomplete group
        await buf.add_episode(make_episode("t2", 0))
        await buf.add_episode(make_episode("t1", 1, wv=4))
        assert buf.qsize() == 1  # t1 complete
        batches = await buf.get_batches(1)
        assert batches[0].task_id == "t1"
        assert len(batches[0].groups) == 1
        assert len(batches[0].groups[0].trajectories) == 2
        assert sorted(batches[0].weight_versions) == [3, 4]
        # offload dir is cleaned after restore
        assert list((tmp_path / "off").glob("*.pkl")) == []

    asyncio.run(run())


@rllm_amd.rollout
def async_flow(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user", "content": str(task.instruction)}]},
                   timeout=30.0)
    r.raise_for_status()
    return None


@rllm_amd.evaluator
def async_eval(task, episode):
    step = episode.trajectories[0].steps[-1]
    if not step.response_ids:
        return 0.0
    return float(sum(1 for t in step.response_ids if t % 2 == 0) / len(step.response_ids))


@pytest.mark.timeout(300)
def test_fully_async_trainer_end_to_end(tmp_path):
    backend = CPUBackend(async_flow, async_eval, rollout_max_tokens=6, seed=0)
    ds = Dataset([{"question": f"q{i}", "id": str(i)} for i in range(6)]).as_tasks(id_key="id")
    cfg = TrainerConfig(total_epochs=1, train_batch_size=2, rollout_n=2, max_steps=3,
                        logger_backends=[])
    acfg = AsyncTrainingConfig(enable=True, mini_batch_size=1, staleness_threshold=1.0,
                               trigger_parameter_sync_step=1, partial_rollout=False)
    trainer = FullyAsyncTrainer(backend, ds, config=cfg, async_config=acfg)
    trainer.fit()
    assert trainer.state.global_step >= 1
    assert trainer.state.weight_version >= 1
