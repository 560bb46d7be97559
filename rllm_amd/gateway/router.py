"""Session → worker routing (reference: session_router.py:25-105).

StickyLeastLoadedPolicy: a session stays on the worker that served it
first (KV/prefix locality on the inference side); new sessions go to the
least-loaded healthy worker. This IS the DP-of-inference layer.
"""

from __future__ import annotations

import time
from collections import OrderedDict
from typing import Protocol

from rllm_amd.gateway.models import WorkerConfig, WorkerInfo


class RoutingPolicy(Protocol):
    def pick(self, session_id: str | None, workers: list[WorkerInfo]) -> WorkerInfo: ...
    def release_session(self, session_id: str) -> None: ...


class StickyLeastLoadedPolicy:
    def __init__(self, max_sessions: int = 100_000):
        self._session_worker: OrderedDict[str, str] = OrderedDict()
        self._max_sessions = max_sessions

    def pick(self, session_id: str | None, workers: list[WorkerInfo]) -> WorkerInfo:
        healthy = [w for w in workers if w.healthy]
        if not healthy:
            raise RuntimeError("no healthy workers registered")
        if session_id is not None:
            url = self._session_worker.get(session_id)
            if url is not None:
                self._session_worker.move_to_end(session_id)
                for w in healthy:
                    if w.url == url:
                        return w
                # sticky worker went unhealthy — fall through to least-loaded
        w = min(healthy, key=lambda w: (w.active_requests, w.active_sessions, w.total_requests))
        if session_id is not None:
            self._session_worker[session_id] = w.url
            w.active_sessions += 1
            while len(self._session_worker) > self._max_sessions:
                self._session_worker.popitem(last=False)
        return w

    def release_session(self, session_id: str) -> None:
        self._session_worker.pop(session_id, None)


class WorkerRegistry:
    """Tracks workers + health; used by the proxy and the admin API."""

    def __init__(self):
        self.workers: dict[str, WorkerInfo] = {}
        self.last_health_check: float = 0.0

    def add(self, cfg: WorkerConfig) -> WorkerInfo:
        info = self.workers.get(cfg.url)
        if info is None:
            info = WorkerInfo(url=cfg.url, model=cfg.model)
            self.workers[cfg.url] = info
        return info

    def remove(self, url: str) -> bool:
        return self.workers.pop(url, None) is not None

    def list(self) -> list[WorkerInfo]:
        return list(self.workers.values())

    def mark(self, url: str, healthy: bool):
        if url in self.workers:
            self.workers[url].healthy = healthy

    async def health_check(self, client) -> None:
        """Probe /health on each worker; drop unhealthy ones from routing."""
        for w in list(self.workers.values()):
            try:
                r = await client.get(w.url.rstrip("/") + "/health", timeout=5.0)
                w.healthy = r.status_code == 200
            except Exception:  # noqa: BLE001
                w.healthy = False
        self.last_health_check = time.time()
