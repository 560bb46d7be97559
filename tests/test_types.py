"""Core data model tests: serde round-trips, coercion, conventions."""

import json

import pytest

from rllm_amd.engine.rollout.model_output import ModelOutput
from rllm_amd.types import (
    AgentConfig,
    Episode,
    Step,
    Task,
    Trajectory,
    TrajectoryGroup,
    _coerce_to_episode,
    run_agent_flow,
)
from rllm_amd.workflows.workflow import TerminationReason


def make_step(reward=0.0, n_tokens=3):
    return Step(
        prompt_ids=[1, 2, 3],
        response_ids=list(range(n_tokens)),
        logprobs=[-0.5] * n_tokens,
        chat_completions=[{"role": "user", "content": "hi"}],
        model_response="yo",
        reward=reward,
        done=True,
    )


def test_step_serde_roundtrip():
    step = make_step(reward=1.0)
    d = step.to_dict()
    # JSON-serializable
    json.dumps(d)
    step2 = Step.from_dict(d)
    assert step2.prompt_ids == step.prompt_ids
    assert step2.response_ids == step.response_ids
    assert step2.logprobs == step.logprobs
    assert step2.reward == 1.0
    assert step2.done is True


def test_step_logprob_length_mismatch_raises():
    with pytest.raises(ValueError):
        Step(response_ids=[1, 2], logprobs=[-0.1])


def test_step_backfills_from_model_output():
    mo = ModelOutput(
        content="answer",
        prompt_ids=[5, 6],
        completion_ids=[7, 8, 9],
        logprobs=[-0.1, -0.2, -0.3],
        weight_version=4,
    )
    step = Step(model_output=mo)
    assert step.prompt_ids == [5, 6]
    assert step.response_ids == [7, 8, 9]
    assert step.logprobs == [-0.1, -0.2, -0.3]
    assert step.weight_version == 4


def test_step_from_model_output_builds_chat():
    mo = ModelOutput(content="c", reasoning="r", prompt_ids=[1], completion_ids=[2], logprobs=[-1.0])
    step = Step.from_model_output(mo, messages=[{"role": "user", "content": "q"}])
    assert step.chat_completions[-1]["role"] == "assistant"
    assert step.thought == "r"
    assert step.model_response == "c"


def test_episode_serde_roundtrip():
    traj = Trajectory(name="solver", steps=[make_step(reward=1.0)], reward=1.0)
    ep = Episode(
        id="task42:3",
        task={"question": "2+2"},
        termination_reason=TerminationReason.ENV_DONE,
        is_correct=True,
        trajectories=[traj],
    )
    d = ep.to_dict()
    json.dumps(d)
    assert d["termination_reason"] == "env_done"
    ep2 = Episode.from_dict(d)
    assert ep2.id == "task42:3"
    assert ep2.task_id == "task42"
    assert ep2.rollout_idx == "3"
    assert ep2.is_correct
    assert ep2.termination_reason == TerminationReason.ENV_DONE
    assert len(ep2.trajectories) == 1
    assert ep2.trajectories[0].steps[0].reward == 1.0


def test_episode_task_sanitizes_images():
    ep = Episode(id="t:0", task={"q": 1, "image": b"\x00", "images": ["x"]})
    d = ep.to_dict()
    assert "image" not in d["task"] and "images" not in d["task"]


def test_trajectory_is_cumulative():
    s1 = Step(chat_completions=[{"role": "user", "content": "a"}])
    s2 = Step(chat_completions=[{"role": "user", "content": "a"}, {"role": "assistant", "content": "b"}])
    t = Trajectory(steps=[s1, s2])
    assert t.is_cumulative()
    s3 = Step(chat_completions=[{"role": "user", "content": "DIFFERENT"}])
    t2 = Trajectory(steps=[s1, s3])
    assert not t2.is_cumulative()


def test_trajectory_group_conventions():
    g = TrajectoryGroup(trajectories=[], group_id="task7:judge")
    assert g.task_id == "task7"
    assert g.group_role == "judge"


def test_coerce_to_episode():
    task = Task(id="t", instruction="do it", metadata={"gt": 4})
    ep = _coerce_to_episode(None, task, "solver")
    assert len(ep.trajectories) == 1
    assert ep.trajectories[0].name == "solver"
    assert ep.task == {"gt": 4}

    traj = Trajectory()
    ep2 = _coerce_to_episode(traj, task, "judge")
    assert ep2.trajectories[0].name == "judge"

    with pytest.raises(TypeError):
        _coerce_to_episode(42, task, "x")


@pytest.mark.parametrize("is_async", [False, True])
def test_run_agent_flow(is_async):
    import asyncio

    task = Task(id="t", instruction="hi")
    config = AgentConfig(base_url="http://x", model="m", session_uid="t:0")

    if is_async:

        class Flow:
            name = "af"

            async def arun(self, task, config):
                return Trajectory(output="done")

    else:

        class Flow:
            name = "sf"

            def run(self, task, config):
                return Trajectory(output="done")

    ep = asyncio.run(run_agent_flow(Flow(), task, config))
    assert ep.trajectories[0].output == "done"
