"""CPU tests for the Episode -> packed training batch transform
(prefix-merge + response-mask semantics, the subtle part per SURVEY.md §7)."""

import pytest
import torch

from rllm_amd.trainer.batch import (
    pack_rows,
    rows_from_episodes,
    rows_from_trajectory,
    split_rows_token_balanced,
)
from rllm_amd.types import Episode, Step, Trajectory


def test_single_step_row():
    st = Step(prompt_ids=[1, 2, 3], response_ids=[10, 11, 12], logprobs=[-0.1, -0.2, -0.3],
              chat_completions=[{"role": "user", "content": "a"}], advantage=0.5)
    rows = rows_from_trajectory(Trajectory(name="s", steps=[st], reward=1.0))
    assert len(rows) == 1
    r = rows[0]
    assert r.tokens == [1, 2, 3, 10, 11, 12]
    assert r.response_mask == [0, 0, 0, 1, 1, 1]
    assert r.advantages == [0.0, 0.0, 0.0, 0.5, 0.5, 0.5]
    assert r.rollout_logprobs == [0.0, 0.0, 0.0, -0.1, -0.2, -0.3]


def test_cumulative_prefix_merge():
    s1 = Step(prompt_ids=[1, 2, 3], response_ids=[10, 11], logprobs=[-0.1, -0.2],
              chat_completions=[{"role": "user", "content": "a"}], advantage=1.0)
    s2 = Step(prompt_ids=[1, 2, 3, 10, 11, 4, 5], response_ids=[12, 13], logprobs=[-0.3, -0.4],
              chat_completions=[{"role": "user", "content": "a"}, {"role": "assistant", "content": "b"}],
              advantage=1.0)
    rows = rows_from_trajectory(Trajectory(name="s", steps=[s1, s2], reward=1.0))
    assert len(rows) == 1
    row = rows[0]
    assert row.tokens == [1, 2, 3, 10, 11, 4, 5, 12, 13]
    assert row.response_mask == [0, 0, 0, 1, 1, 0, 0, 1, 1]
    assert row.rollout_logprobs == [0.0, 0.0, 0.0, -0.1, -0.2, 0.0, 0.0, -0.3, -0.4]


def test_non_cumulative_falls_back_to_per_step_rows():
    s1 = Step(prompt_ids=[1, 2], response_ids=[10], logprobs=[-0.1],
              chat_completions=[{"role": "user", "content": "a"}])
    s2 = Step(prompt_ids=[9, 9, 9], response_ids=[12], logprobs=[-0.3],
              chat_completions=[{"role": "user", "content": "a"}, {"role": "assistant", "content": "b"}])
    # chat_completions look cumulative but token prefix does NOT match:
    rows = rows_from_trajectory(Trajectory(name="s", steps=[s1, s2], reward=1.0))
    assert len(rows) == 2


def test_pack_rows_shift_semantics():
    st = Step(prompt_ids=[1, 2, 3], response_ids=[10, 11], logprobs=[-0.5, -0.6],
              chat_completions=[{"role": "user", "content": "a"}], advantage=2.0)
    rows = rows_from_trajectory(Trajectory(name="s", steps=[st], reward=1.0))
    batch = pack_rows(rows)
    # logprob of token t is computed at row t-1:
    # rows 0..4 = [1,2,3,10,11]; loss rows = 2 (target 10), 3 (target 11)
    assert batch.loss_mask.tolist() == [False, False, True, True, False]
    assert batch.targets[batch.loss_mask].tolist() == [10, 11]
    assert batch.advantages[batch.loss_mask].tolist() == [2.0, 2.0]
    assert batch.rollout_logprobs[batch.loss_mask].tolist() == pytest.approx([-0.5, -0.6])
    assert batch.n_response_tokens == 2
    assert batch.cu_seqlens == [0, 5]


def test_pack_rows_multiple_sequences():
    eps = []
    for i in range(3):
        st = Step(prompt_ids=[1] * (i + 2), response_ids=[5, 6], logprobs=[-0.1, -0.1],
                  chat_completions=[{"role": "user", "content": "a"}], advantage=1.0)
        eps.append(Episode(id=f"t{i}:0", trajectories=[Trajectory(name="s", steps=[st], reward=1.0)]))
    rows = rows_from_episodes(eps)
    batch = pack_rows(rows)
    assert batch.n_rows == 3
    assert batch.n_response_tokens == 6
    assert batch.cu_seqlens[-1] == batch.input_ids.numel()
    # positions restart at 0 per sequence
    pos = batch.positions.tolist()
    for b in range(3):
        s0, s1 = batch.cu_seqlens[b], batch.cu_seqlens[b + 1]
        assert pos[s0:s1] == list(range(s1 - s0))


def test_token_balanced_split():
    sts = []
    for n in [100, 90, 50, 40, 30, 10]:
        sts.append(Step(prompt_ids=[1] * (n - 5), response_ids=[2] * 5, logprobs=[-0.1] * 5,
                        chat_completions=[{"role": "user", "content": "a"}]))
    rows = [rows_from_trajectory(Trajectory(name="s", steps=[s], reward=0.0))[0] for s in sts]
    micros = split_rows_token_balanced(rows, 128)
    assert all(sum(len(r) for r in m) <= 128 for m in micros)
    assert sum(len(m) for m in micros) == 6


def test_max_seq_len_truncation():
    st = Step(prompt_ids=[1] * 10, response_ids=[2] * 10, logprobs=[-0.1] * 10,
              chat_completions=[{"role": "user", "content": "a"}], advantage=1.0)
    rows = rows_from_trajectory(Trajectory(name="s", steps=[st], reward=1.0))
    batch = pack_rows(rows, max_seq_len=15)
    assert batch.input_ids.numel() == 15
    assert batch.n_response_tokens == 5  # tokens 10..14 are responses (shifted)


def test_pack_rows_pad_to_multiple():
    from rllm_amd.trainer.batch import PackedRow, pack_rows

    rows = [PackedRow(tokens=[1, 2, 3, 4, 5], response_mask=[0, 0, 1, 1, 1],
                      advantages=[0.5] * 5, rollout_logprobs=[-1.0] * 5)]
    for P in (2, 3, 4, 8):
        b = pack_rows(rows, pad_to_multiple=P)
        assert b.input_ids.shape[0] % P == 0, P
        # pad tokens are never loss rows and never counted
        assert b.n_response_tokens == 3
        assert not b.loss_mask[b.cu_seqlens[-2]:].any()
    # already divisible: no pad sequence added
    b = pack_rows(rows, pad_to_multiple=5)
    assert b.input_ids.shape[0] == 5 and b.n_rows == 1
