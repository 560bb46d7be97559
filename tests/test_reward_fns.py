"""MCQ / F1 / countdown / format reward graders."""

import pytest

from rllm_amd.rewards.reward_fns import (
    countdown_reward,
    exact_match_reward,
    extract_mcq_choice,
    f1_reward,
    format_reward,
    mcq_reward,
)


def test_mcq_extraction():
    assert extract_mcq_choice("The answer is (C)") == "C"
    assert extract_mcq_choice("answer: b") == "B"
    assert extract_mcq_choice(r"so \boxed{D}") == "D"
    assert extract_mcq_choice("no choice here") is None


def test_mcq_reward():
    assert mcq_reward("I pick (A) because...the answer is A", "a") == 1.0
    assert mcq_reward("the answer is B", "A") == 0.0


def test_f1():
    assert f1_reward("the cat sat", "cat sat") == 1.0
    assert f1_reward("dog", "cat") == 0.0
    assert 0 < f1_reward("the red cat", "blue cat") < 1


def test_exact_match():
    assert exact_match_reward("  Paris ", "paris") == 1.0
    assert exact_match_reward("London", "paris") == 0.0


def test_countdown():
    assert countdown_reward(r"\boxed{(3+5)*2}", 16, [3, 5, 2]) == 1.0
    assert countdown_reward(r"\boxed{(3+5)*2}", 17, [3, 5, 2]) == 0.0  # wrong value
    assert countdown_reward(r"\boxed{3*3+5}", 14, [3, 5]) == 0.0  # reuses 3
    assert countdown_reward("import os", 1, [1]) == 0.0  # charset rejected


def test_format_reward():
    assert format_reward(r"\boxed{4}", require_boxed=True) == 1.0
    assert format_reward("plain", require_boxed=True) == 0.0
    assert format_reward("<think>hm</think> yes", require_think=True, max_words=10) == 1.0
    assert format_reward("a " * 50, max_words=10) == 0.0
