"""End-to-end CPU integration: @rollout flow → gateway session → fake
worker → trace capture → enrichment → @evaluator → rewards (mirrors the
reference's test_training_workflow_e2e.py)."""

import asyncio
import sys
from pathlib import Path

import httpx
import pytest

sys.path.insert(0, str(Path(__file__).parent))

from helpers.fake_worker import FakeWorkerServer  # noqa: E402

import rllm_amd  # noqa: E402
from rllm_amd.engine.agentflow_engine import AgentFlowEngine, EnrichMismatchError, enrich_episode_with_traces  # noqa: E402
from rllm_amd.gateway.manager import GatewayManager  # noqa: E402
from rllm_amd.gateway.models import GatewayConfig, TraceRecord  # noqa: E402
from rllm_amd.types import Episode, Step, Task, Trajectory  # noqa: E402


@pytest.fixture(scope="module")
def stack():
    with FakeWorkerServer() as worker:
        gw = GatewayManager(GatewayConfig())
        gw.start(worker_urls=[worker.url])
        yield gw, worker
        gw.stop()


@rllm_amd.rollout
def math_flow(task, config):
    # plain OpenAI-style call through the gateway session URL
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user", "content": task.instruction}]},
                   timeout=30.0)
    r.raise_for_status()
    return None  # framework builds the Episode from traces


@rllm_amd.evaluator
def exact_match(task, episode):
    # fake worker always answers "fake response"
    resp = episode.trajectories[0].steps[-1].model_response
    return 1.0 if "fake" in resp else 0.0


def test_full_rollout_pipeline(stack):
    gw, worker = stack
    engine = AgentFlowEngine(math_flow, gw, evaluator=exact_match, n_parallel_tasks=8)
    tasks = [Task(id=f"t{i}", instruction=f"what is {i}+{i}?", metadata={"gt": 2 * i}) for i in range(4)]
    task_ids = [f"t{i}:0" for i in range(4)]
    episodes = asyncio.run(engine.execute_tasks(tasks, task_ids))

    assert len(episodes) == 4
    for ep, tid in zip(episodes, task_ids):
        assert ep.id == tid
        assert ep.is_correct
        traj = ep.trajectories[0]
        assert traj.reward == 1.0
        step = traj.steps[0]
        # training payloads filled from gateway traces
        assert step.response_ids == [7, 8, 9]
        assert step.logprobs == pytest.approx([-0.11, -0.22, -0.33])
        assert len(step.prompt_ids) > 0
        assert step.weight_version == 3
        assert step.reward == 1.0  # written onto last step
        assert "response_tokens" in ep.metrics
        assert "time/agentflow_s" in ep.metrics

    # sessions were batch-deleted
    assert gw.client().get_traces(task_ids[0]) == []


def test_rollout_n_interleave(stack):
    """rollout.n style: same task, multiple rollout uids."""
    gw, _ = stack
    engine = AgentFlowEngine(math_flow, gw, evaluator=exact_match)
    task = Task(id="taskX", instruction="solve")
    uids = [f"taskX:{i}" for i in range(3)]
    episodes = asyncio.run(engine.execute_tasks([task] * 3, uids))
    assert [e.id for e in episodes] == uids
    assert all(e.task_id == "taskX" for e in episodes)


def test_enrich_strict_raises_on_empty_token_ids():
    traces = [TraceRecord(session_id="u:0", messages=[], prompt_token_ids=[],
                          completion_token_ids=[], logprobs=[])]
    ep = Episode(id="u:0", trajectories=[])
    with pytest.raises(EnrichMismatchError):
        enrich_episode_with_traces(ep, traces, "u:0", {}, strict=True)
    # relaxed (eval) path accepts
    out = enrich_episode_with_traces(ep, traces, "u:0", {}, strict=False)
    assert len(out.trajectories) == 1


def test_enrich_drops_trailing_malformed_trace():
    good = TraceRecord(session_id="u:0", prompt_token_ids=[1, 2], completion_token_ids=[3],
                       logprobs=[-0.1], response_message={"role": "assistant", "content": "x"})
    bad = TraceRecord(session_id="u:0", prompt_token_ids=[], completion_token_ids=[])
    agent_step = Step(chat_completions=[{"role": "user", "content": "q"}], reward=0.7, done=True)
    ep = Episode(id="u:0", trajectories=[Trajectory(name="s", steps=[agent_step])])
    out = enrich_episode_with_traces(ep, [good, bad], "u:0", {}, strict=True)
    assert len(out.trajectories[0].steps) == 1
    st = out.trajectories[0].steps[0]
    assert st.response_ids == [3]
    assert st.reward == 0.7  # agent-side fields preserved


def test_error_becomes_error_episode(stack):
    gw, _ = stack

    @rllm_amd.rollout
    def broken_flow(task, config):
        raise RuntimeError("boom")

    engine = AgentFlowEngine(broken_flow, gw, max_retries=2)
    eps = asyncio.run(engine.execute_tasks([Task(id="e", instruction="x")], ["e:0"]))
    from rllm_amd.workflows.workflow import TerminationReason
    assert eps[0].termination_reason == TerminationReason.ERROR
    assert "boom" in eps[0].info["error"]["error_message"]
