"""Math reward grading tests (boxed extraction, normalization, sympy)."""

import pytest

from rllm_amd.rewards.math_reward import (
    RewardMathFn,
    extract_answer,
    extract_boxed,
    grade_answer,
    normalize_answer,
)


def test_extract_boxed_balanced():
    assert extract_boxed(r"So the answer is \boxed{42}.") == "42"
    assert extract_boxed(r"\boxed{\frac{1}{2}}") == r"\frac{1}{2}"
    assert extract_boxed(r"first \boxed{1} then \boxed{2}") == "2"
    assert extract_boxed("no box") is None


def test_extract_answer_fallbacks():
    assert extract_answer("The final answer: 17") == "17"
    assert extract_answer("blah 3 blah 7.5") == "7.5"
    assert extract_answer(r"thus \boxed{x+1}") == "x+1"


@pytest.mark.parametrize("a,b", [
    ("42", "42"),
    (r"\frac{1}{2}", "1/2"),
    ("0.5", "1/2"),
    (r"\frac{1}{2}", "0.5"),
    ("  42. ", "42"),
    (r"\$3", "3"),
    ("50\\%", "50"),
    (r"2\cdot 3", "6"),
    ("sqrt(4)", "2"),
    (r"\sqrt{4}", "2"),
    ("x = 5", "5"),
])
def test_grade_equal(a, b):
    assert grade_answer(a, b), (normalize_answer(a), normalize_answer(b))


@pytest.mark.parametrize("a,b", [
    ("42", "41"),
    ("1/2", "1/3"),
    ("x+1", "x+2"),
    (None, "5"),
])
def test_grade_unequal(a, b):
    assert not grade_answer(a, b)


def test_reward_fn():
    fn = RewardMathFn()
    out = fn(r"Let me think. The answer is \boxed{\frac{2}{4}}", "1/2")
    assert out.is_correct and out.reward == 1.0
    out = fn("I don't know", "7")
    assert not out.is_correct and out.reward == 0.0
