"""RemoteAgentFlowEngine against a stub runtime: the 'remote agent' chats
through the gateway session URL, the engine assembles the Episode from
traces and merges the runtime's reward."""

import asyncio

import httpx
import pytest
from fastapi import FastAPI

from rllm_amd.engine.remote_agentflow_engine import RemoteAgentFlowEngine, episode_from_traces
from rllm_amd.engine.remote_runtime import HttpRemoteRuntime
from rllm_amd.gateway.manager import GatewayManager
from rllm_amd.gateway.models import GatewayConfig
from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
from rllm_amd.models.torch_lm import TinyTorchLM
from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
from rllm_amd.types import Task
from rllm_amd.workflows.workflow import TerminationReason
from rllm_amd.utils.tokenizer import ByteTokenizer


def _stub_runtime_app() -> FastAPI:
    """Fake remote runtime: on submit, act as the remote agent (one chat
    call against the gateway session), then report done with a reward."""
    app = FastAPI()
    runs: dict[str, dict] = {}

    @app.post("/runs")
    async def submit(body: dict):
        rid = body["run_id"]
        if body["metadata"].get("explode"):
            runs[rid] = {"status": "failed", "error": "sandbox crashed"}
            return {"run_id": rid}
        async with httpx.AsyncClient(timeout=60.0) as client:
            r = await client.post(body["gateway_url"] + "/chat/completions",
                                  json={"model": body["model"],
                                        "messages": [{"role": "user",
                                                      "content": body["instruction"]}],
                                        "max_tokens": 4})
            r.raise_for_status()
        runs[rid] = {"status": "done", "reward": 0.75,
                     "metadata": {"runtime": "stub", "turns": 1}}
        return {"run_id": rid}

    @app.get("/runs/{rid}")
    async def status(rid: str):
        return runs.get(rid, {"status": "running"})

    return app


@pytest.fixture()
def gw():
    parser = QwenChatTemplateParser(ByteTokenizer())
    handler = make_torch_lm_local_handler(TinyTorchLM(seed=0), parser)
    g = GatewayManager(GatewayConfig(), local_handler=handler, parser=parser)
    g.start()
    yield g
    g.stop()


def _engine(gw):
    transport = httpx.ASGITransport(app=_stub_runtime_app())
    runtime = HttpRemoteRuntime(
        "http://runtime", poll_interval=0.05, timeout=30.0,
        client=httpx.AsyncClient(transport=transport, base_url="http://runtime"))
    return RemoteAgentFlowEngine(runtime, gw, model_name="m",
                                 default_sampling_params={"max_tokens": 4},
                                 n_parallel_tasks=4, max_retries=1)


def test_remote_run_builds_episode_from_traces(gw):
    eng = _engine(gw)
    tasks = [Task(id="r0", instruction="add 2 and 2", metadata={"answer": "4"})]
    eps = asyncio.run(eng.execute_tasks(tasks, ["r0:0"]))
    ep = eps[0]
    assert ep.termination_reason == TerminationReason.ENV_DONE
    assert len(ep.trajectories) == 1
    step = ep.trajectories[0].steps[0]
    assert step.response_ids and step.logprobs          # token traces captured
    assert ep.trajectories[0].reward == 0.75            # remote reward merged
    assert ep.is_correct
    assert ep.info["remote"]["runtime"] == "stub"


def test_remote_failure_is_error_episode(gw):
    eng = _engine(gw)
    tasks = [Task(id="r1", instruction="boom", metadata={"explode": True})]
    eps = asyncio.run(eng.execute_tasks(tasks, ["r1:0"]))
    ep = eps[0]
    assert ep.termination_reason == TerminationReason.ERROR
    assert "error" in ep.info


def test_episode_from_traces_empty():
    ep = episode_from_traces("x:0", {}, [])
    assert ep.trajectories == []
