"""Eval episode store, curation + filter DSL."""

import pytest

from rllm_amd.eval.curation import curate, episodes_to_sft_rows, filter_episodes, parse_filter
from rllm_amd.eval.episode_store import EvalEpisodeStore
from rllm_amd.types import Episode, Step, Trajectory
from rllm_amd.workflows.workflow import TerminationReason


def make_ep(tid, idx, reward, correct, n_tokens=3):
    st = Step(prompt_ids=[1, 2], response_ids=list(range(n_tokens)), logprobs=[-0.1] * n_tokens,
              chat_completions=[{"role": "user", "content": "q"},
                                {"role": "assistant", "content": f"a{tid}"}],
              reward=reward, done=True)
    return Episode(id=f"{tid}:{idx}", is_correct=correct,
                   termination_reason=TerminationReason.ENV_DONE,
                   trajectories=[Trajectory(name="s", steps=[st], reward=reward)])


def test_episode_store_roundtrip(tmp_path):
    store = EvalEpisodeStore(tmp_path)
    eps = [make_ep("a", 0, 1.0, True), make_ep("b", 0, 0.0, False)]
    store.save_run("run1", eps, metrics={"accuracy": 0.5})
    assert store.list_runs()["run1"]["num_episodes"] == 2
    loaded = store.load_run("run1")
    assert len(loaded) == 2
    assert {e.task_id for e in loaded} == {"a", "b"}
    store.delete_run("run1")
    assert "run1" not in store.list_runs()


def test_filter_dsl():
    eps = [make_ep("a", 0, 1.0, True), make_ep("b", 0, 0.2, False),
           make_ep("c", 0, 0.8, True, n_tokens=10)]
    assert len(filter_episodes(eps, ["is_correct == True"])) == 2
    assert len(filter_episodes(eps, ["reward >= 0.8"])) == 2
    assert len(filter_episodes(eps, ["response_tokens > 5"])) == 1
    assert len(filter_episodes(eps, ["termination_reason == env_done"])) == 3
    assert len(filter_episodes(eps, ["is_correct == True", "reward < 0.9"])) == 1
    with pytest.raises(ValueError):
        parse_filter("not a filter")


def test_curation_to_sft_rows():
    eps = [make_ep("a", 0, 1.0, True), make_ep("a", 1, 0.5, True), make_ep("b", 0, 0.0, False)]
    rows = curate(eps, filters=["is_correct == True"], top_k_per_task=1)
    assert len(rows) == 1
    assert rows[0]["task_id"] == "a" and rows[0]["reward"] == 1.0
    assert rows[0]["messages"][-1]["role"] == "assistant"

    # dedupe identical transcripts
    rows2 = episodes_to_sft_rows([make_ep("x", 0, 1.0, True), make_ep("x", 1, 1.0, True)])
    assert len(rows2) == 1
