"""LLM-as-judge reward: parsing, retry/fallback, RewardFn protocol."""

import pytest

from rllm_amd.rewards.llm_judge import LLMJudgeReward, parse_judge_score
from rllm_amd.types import Episode, Step, Task, Trajectory


@pytest.mark.parametrize("text,expect", [
    ('{"score": 1.0}', 1.0),
    ('Sure! {"score": 0.5} because...', 0.5),
    ("score: 0.75", 0.75),
    ("The answer is correct.", 1.0),
    ("That is incorrect.", 0.0),
    ('{"score": 7}', 1.0),       # clamped
    ("no json here at all ???", 0.0),  # bare "no" verdict
    ("", None),
    ("meaningless gibberish", None),
])
def test_parse_judge_score(text, expect):
    assert parse_judge_score(text) == expect


def _episode(answer: str) -> Episode:
    return Episode(id="e0", task={}, trajectories=[
        Trajectory(name="t", steps=[Step(model_response=answer)])])


def test_judge_grades_via_chat_fn():
    calls = []

    def chat(messages):
        calls.append(messages)
        assert messages[0]["role"] == "system"
        assert "2+2" in messages[1]["content"]
        return '{"score": 1.0}' if "4" in messages[1]["content"] else '{"score": 0}'

    judge = LLMJudgeReward(chat_fn=chat)
    task = Task(id="t0", instruction="What is 2+2?", metadata={"answer": "4"})
    assert judge(task, _episode("It is 4.")) == 1.0
    assert len(calls) == 1


def test_judge_retry_then_default():
    replies = iter(["asdf qwerty zzz", "more gibberish zzz"])
    judge = LLMJudgeReward(chat_fn=lambda m: next(replies),
                           default_score=0.25, retries=1)
    task = Task(id="t", instruction="q", metadata={"answer": "a"})
    assert judge(task, _episode("something")) == 0.25


def test_judge_endpoint_failure_is_default():
    def chat(messages):
        raise ConnectionError("down")

    judge = LLMJudgeReward(chat_fn=chat, default_score=0.0)
    task = Task(id="t", instruction="q", metadata={"answer": "a"})
    assert judge(task, _episode("x")) == 0.0


def test_empty_candidate_short_circuits():
    judge = LLMJudgeReward(chat_fn=lambda m: pytest.fail("must not call"))
    task = Task(id="t", instruction="q", metadata={"answer": "a"})
    assert judge(task, _episode("")) == 0.0


def test_requires_some_endpoint():
    with pytest.raises(ValueError):
        LLMJudgeReward()
