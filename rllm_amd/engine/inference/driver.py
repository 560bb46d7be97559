"""AsyncEngineDriver: owns the LLMEngine on a dedicated thread and exposes
an awaitable submit() API. The engine thread runs the continuous-batching
loop; completions resolve asyncio futures on their home loops.

This is the in-process bridge used both by the OpenAI-compatible server
and by the NativeEngine rollout adapter (no socket hop when colocated).
"""

from __future__ import annotations

import asyncio
import queue
import threading
import uuid

from rllm_amd.engine.inference.llm_engine import LLMEngine, RequestOutput, SamplingParams


class AsyncEngineDriver:
    def __init__(self, engine: LLMEngine):
        self.engine = engine
        self._submit_q: queue.SimpleQueue = queue.SimpleQueue()
        self._futures: dict[str, tuple[asyncio.AbstractEventLoop, asyncio.Future]] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True, name="llm-engine")
        self._thread.start()

    # ------------------------------------------------------------------
    async def submit(self, prompt_ids: list[int], params: SamplingParams,
                     request_id: str | None = None) -> RequestOutput:
        rid = request_id or str(uuid.uuid4())
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        with self._lock:
            self._futures[rid] = (loop, fut)
        self._submit_q.put(("add", rid, prompt_ids, params))
        self._wake.set()
        return await fut

    def abort(self, request_id: str) -> None:
        self._submit_q.put(("abort", request_id, None, None))
        self._wake.set()

    def pause(self):
        self._submit_q.put(("pause", None, None, None))
        self._wake.set()

    def resume(self):
        self._submit_q.put(("resume", None, None, None))
        self._wake.set()

    def set_weight_version(self, v: int):
        self._submit_q.put(("weight_version", v, None, None))
        self._wake.set()

    def drain_wait(self, timeout: float = 600.0) -> bool:
        """Block until no in-flight work (weight-sync drain)."""
        import time

        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if not self.engine.has_unfinished() and self._submit_q.empty():
                return True
            import time as _t

            _t.sleep(0.01)
        return False

    def shutdown(self):
        self._stop.set()
        self._wake.set()
        self._thread.join(timeout=10)

    # ------------------------------------------------------------------
    def _loop(self):
        eng = self.engine
        while not self._stop.is_set():
            # drain control queue
            while True:
                try:
                    op, a, b, c = self._submit_q.get_nowait()
                except queue.Empty:
                    break
                if op == "add":
                    eng.add_request(a, b, c)
                elif op == "abort":
                    eng.abort(a)
                elif op == "pause":
                    eng.pause()
                elif op == "resume":
                    eng.resume()
                elif op == "weight_version":
                    eng.weight_version = a

            if eng.has_unfinished():
                eng.step()
                for out in eng.pop_finished():
                    with self._lock:
                        entry = self._futures.pop(out.request_id, None)
                    if entry is not None:
                        loop, fut = entry
                        loop.call_soon_threadsafe(
                            lambda f=fut, o=out: (not f.cancelled()) and f.set_result(o))
            else:
                # idle: wait for work
                self._wake.wait(timeout=0.05)
                self._wake.clear()
