"""Config assembly, metrics aggregator, harness layer, tokenizer edge cases."""

import asyncio

import pytest

from rllm_amd.config import apply_override, load_config, select
from rllm_amd.harnesses.cli_harness import BashHarness, CurlChatHarness, get_harness
from rllm_amd.sandbox.backends import LocalSandbox
from rllm_amd.trainer.metrics_aggregator import MetricsAggregator
from rllm_amd.types import AgentConfig, Task


def test_load_config_defaults_match_reference():
    cfg = load_config()
    assert cfg["rollout"]["n"] == 8
    assert cfg["rollout"]["temperature"] == 1.0
    assert cfg["algorithm"]["eps_clip"] == 0.2
    assert cfg["algorithm"]["kl_beta"] == 0.001
    assert cfg["actor"]["lr"] == 1e-6
    assert cfg["data"]["max_response_length"] == 30720


def test_config_overrides_and_select():
    cfg = load_config(overrides=["rollout.n=4", "algorithm.estimator=rloo",
                                 "trainer.logger=[console,jsonl]"])
    assert cfg["rollout"]["n"] == 4
    assert cfg["algorithm"]["estimator"] == "rloo"
    assert cfg["trainer"]["logger"] == ["console", "jsonl"]
    assert select(cfg, "algorithm.rollout_correction.tis_cap") == 5.0
    assert select(cfg, "no.such.key", 7) == 7
    apply_override(cfg, "new.nested.key", 3)
    assert cfg["new"]["nested"]["key"] == 3


def test_metrics_aggregator():
    agg = MetricsAggregator()
    agg.add({"loss": 1.0, "skip": "str"})
    agg.add({"loss": 3.0})
    agg.count("episodes", 5)
    out = agg.flush()
    assert out["loss"] == 2.0
    assert out["episodes"] == 5
    assert agg.flush() == {}


def test_harness_registry_and_invocation():
    h = get_harness("bash", command_template="echo {instruction}")
    task = Task(id="t", instruction="hello world")
    cfg = AgentConfig(base_url="http://gw/sessions/t:0/v1", model="m", session_uid="t:0")
    cmd = h.build_invocation(task, cfg)
    assert "hello world" in cmd
    env = h.build_env_vars(task, cfg)
    assert env["OPENAI_BASE_URL"] == cfg.base_url
    with pytest.raises(KeyError):
        get_harness("nope")


def test_bash_harness_runs_in_sandbox():
    async def run():
        h = BashHarness(command_template="echo ran-{instruction} > out.txt")
        task = Task(id="t", instruction="ok")
        cfg = AgentConfig(base_url="http://x/v1", model="m", session_uid="t:0")
        sbx = LocalSandbox()
        result = await h.arun(task, cfg, env=sbx)
        assert result is None  # episode comes from traces
        assert (await sbx.read_file("out.txt")).decode().strip() == "ran-ok"
        await sbx.close()

    asyncio.run(run())


def test_curl_chat_harness_invocation_shape():
    h = CurlChatHarness()
    task = Task(id="t", instruction="what is 1+1")
    cfg = AgentConfig(base_url="http://gw/sessions/s/v1", model="m", session_uid="s")
    cmd = h.build_invocation(task, cfg)
    assert "chat/completions" in cmd and "curl" in cmd


def test_ui_logger_posts_batches():
    import threading
    import time as _time

    received = []
    from fastapi import FastAPI
    import uvicorn
    import socket

    app = FastAPI()

    @app.post("/ingest")
    async def ingest(body: dict):
        received.extend(body["events"])
        return {"ok": True}

    @app.post("/heartbeat")
    async def hb(body: dict):
        return {"ok": True}

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                           log_level="warning", access_log=False))
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    import httpx
    for _ in range(100):
        try:
            httpx.get(f"http://127.0.0.1:{port}/docs", timeout=1.0)
            break
        except Exception:
            _time.sleep(0.05)

    from rllm_amd.utils.tracking import UILogger

    ui = UILogger(f"http://127.0.0.1:{port}", run_id="r1", flush_interval=0.3)
    ui.log({"loss": 1.0}, step=0)
    ui.log_episode({"id": "t:0"}, step=0)
    deadline = _time.time() + 8
    while _time.time() < deadline and len(received) < 2:
        _time.sleep(0.1)
    ui.finish()
    server.should_exit = True
    assert len(received) >= 2
    kinds = {e["type"] for e in received}
    assert kinds == {"metrics", "episode"}
