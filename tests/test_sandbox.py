"""Sandbox tests: local backend exec/files, manager, warm queue, and the
SandboxTaskHooks join point."""

import asyncio

import pytest

from rllm_amd.hooks import SandboxTaskHooks
from rllm_amd.sandbox.backends import LocalSandbox, LocalSandboxBackend
from rllm_amd.sandbox.manager import SandboxManager, WarmQueue


def test_local_sandbox_exec_and_files():
    async def run():
        sbx = LocalSandbox()
        r = await sbx.exec("echo hello && echo err >&2")
        assert r.ok and r.stdout.strip() == "hello" and "err" in r.stderr
        await sbx.write_file("sub/a.txt", "content")
        assert (await sbx.read_file("sub/a.txt")) == b"content"
        r2 = await sbx.exec("cat sub/a.txt")
        assert r2.stdout == "content"
        # timeout
        r3 = await sbx.exec("sleep 5", timeout=0.3)
        assert r3.timed_out
        assert await sbx.is_alive()
        await sbx.close()
        assert not await sbx.is_alive()

    asyncio.run(run())


def test_sandbox_manager_concurrency_and_release():
    async def run():
        mgr = SandboxManager(LocalSandboxBackend(), max_concurrent=2)
        a = await mgr.acquire()
        b = await mgr.acquire()
        assert len(mgr.active) == 2
        # third acquire blocks until release
        acquired = asyncio.Event()

        async def third():
            c = await mgr.acquire()
            acquired.set()
            await mgr.release(c)

        t = asyncio.create_task(third())
        await asyncio.sleep(0.05)
        assert not acquired.is_set()
        await mgr.release(a)
        await asyncio.wait_for(acquired.wait(), 3.0)
        await mgr.release(b)
        await t
        assert mgr.active == {}

    asyncio.run(run())


def test_warm_queue_prefetch_and_handoff():
    async def run():
        q = WarmQueue(LocalSandboxBackend(), depth=2)
        await q.start()
        await asyncio.sleep(0.3)  # let it fill
        sbx = await q.acquire()
        assert await sbx.is_alive()
        await sbx.close()
        await q.stop()

    asyncio.run(run())


def test_sandbox_task_hooks_provisions_env():
    async def run():
        mgr = SandboxManager(LocalSandboxBackend())
        hooks = SandboxTaskHooks(sandbox_manager=mgr, fixed_evaluator="EV")
        ctx = await hooks.setup({"id": "t"}, "t:0")
        assert ctx.evaluator == "EV"
        assert ctx.env is not None and await ctx.env.is_alive()
        await hooks.teardown({"id": "t"}, "t:0", ctx)
        assert mgr.active == {}

    asyncio.run(run())


def test_snapshot_registry_dedup_refcount_ttl():
    async def run():
        from rllm_amd.sandbox.snapshot import SnapshotRegistry, snapshot_key

        created, destroyed = [], []

        async def create(spec):
            created.append(spec)
            return f"img-{len(created)}"

        async def destroy(handle):
            destroyed.append(handle)

        reg = SnapshotRegistry(create, destroy, ttl_s=0.0)  # instant expiry when unref'd
        spec = {"image": "py:3.11", "setup": ["pip install x"]}
        k1, h1 = await reg.acquire(spec)
        k2, h2 = await reg.acquire(spec)
        assert k1 == k2 and h1 == h2 and len(created) == 1  # dedup
        k3, _ = await reg.acquire({"image": "other"})
        assert k3 != k1 and len(created) == 2

        await reg.release(k1)
        assert reg.stats()["snapshots"] >= 1  # still referenced once
        await reg.release(k1)
        # both refs gone + ttl 0 -> evicted and destroyed
        await reg.release(k3)
        assert "img-1" in destroyed
        assert snapshot_key(spec) == snapshot_key(dict(reversed(list(spec.items())))), "order-stable hash"

    asyncio.run(run())
