"""A fake inference worker emitting the native engine's wire format
(mirrors the reference's tests/helpers/mock_vllm.py pattern: a FastAPI
fake + a controllable variant with failure injection)."""

from __future__ import annotations

import asyncio
import json
import socket
import threading
import time

import uvicorn
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse


def make_fake_worker_app(model: str = "fake-1.5b", weight_version: int = 3) -> FastAPI:
    app = FastAPI()
    state = {"fail_next": 0, "weight_version": weight_version, "requests": []}
    app.state.ctl = state

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.post("/admin/fail_next")
    async def fail_next(body: dict):
        state["fail_next"] = int(body.get("n", 1))
        return {"ok": True}

    def _gen_payload(body: dict):
        messages = body.get("messages") or []
        n_prompt = 4 + len(messages)
        prompt_ids = list(range(100, 100 + n_prompt))
        completion_ids = [7, 8, 9]
        logprobs = [-0.11, -0.22, -0.33]
        text = "fake response"
        return prompt_ids, completion_ids, logprobs, text

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        body = await request.json()
        state["requests"].append(body)
        if state["fail_next"] > 0:
            state["fail_next"] -= 1
            return JSONResponse({"error": "injected failure"}, status_code=500)
        prompt_ids, completion_ids, logprobs, text = _gen_payload(body)

        if body.get("stream"):
            async def gen():
                for i, (tok, lp) in enumerate(zip(completion_ids, logprobs)):
                    chunk = {
                        "id": "cmpl-1", "object": "chat.completion.chunk", "model": model,
                        "prompt_token_ids": prompt_ids if i == 0 else None,
                        "weight_version": state["weight_version"],
                        "choices": [{
                            "index": 0,
                            "delta": {"content": f"t{i} "},
                            "token_ids": [tok],
                            "logprobs": {"token_logprobs": [lp]},
                            "finish_reason": "stop" if i == len(completion_ids) - 1 else None,
                        }],
                    }
                    yield f"data: {json.dumps(chunk)}\n\n".encode()
                    await asyncio.sleep(0.001)
                yield b"data: [DONE]\n\n"

            return StreamingResponse(gen(), media_type="text/event-stream")

        return {
            "id": "cmpl-1", "object": "chat.completion", "model": model,
            "prompt_token_ids": prompt_ids,
            "weight_version": state["weight_version"],
            "choices": [{
                "index": 0,
                "message": {"role": "assistant", "content": text},
                "token_ids": completion_ids,
                "logprobs": {"token_logprobs": logprobs},
                "finish_reason": "stop",
            }],
            "usage": {"prompt_tokens": len(prompt_ids), "completion_tokens": len(completion_ids)},
        }

    return app


class FakeWorkerServer:
    """Run the fake worker on a background uvicorn thread."""

    def __init__(self, app: FastAPI | None = None):
        self.app = app or make_fake_worker_app()
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            self.port = s.getsockname()[1]
        self.url = f"http://127.0.0.1:{self.port}"
        self._server: uvicorn.Server | None = None
        self._thread: threading.Thread | None = None

    def __enter__(self):
        cfg = uvicorn.Config(self.app, host="127.0.0.1", port=self.port,
                             log_level="warning", access_log=False)
        self._server = uvicorn.Server(cfg)
        self._thread = threading.Thread(target=self._server.run, daemon=True)
        self._thread.start()
        import httpx

        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            try:
                httpx.get(self.url + "/health", timeout=2.0)
                return self
            except Exception:  # noqa: BLE001
                time.sleep(0.05)
        raise RuntimeError("fake worker failed to start")

    def __exit__(self, *exc):
        if self._server:
            self._server.should_exit = True
        if self._thread:
            self._thread.join(timeout=5)
