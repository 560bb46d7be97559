"""Property tests for the training-batch semantics (the safety-critical
trainer plumbing): pack_rows shift alignment, loss-token conservation
across micro splits, and mask/advantage/logprob alignment."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings


def row_strategy():
    @st.composite
    def _row(draw):
        from rllm_amd.trainer.batch import PackedRow

        n = draw(st.integers(2, 40))
        p = draw(st.integers(1, n - 1))
        tokens = draw(st.lists(st.integers(0, 999), min_size=n, max_size=n))
        mask = [0] * p + [1] * (n - p)
        # punch random holes in the response mask (multi-turn observations)
        for i in draw(st.lists(st.integers(0, n - 1), max_size=4)):
            if i >= p:
                mask[i] = 0
        adv = draw(st.floats(-3, 3, allow_nan=False))
        lps = [round(-abs(a) % 7 - 0.25, 3) for a in tokens]
        return PackedRow(tokens=tokens, response_mask=mask,
                         advantages=[adv] * n, rollout_logprobs=lps)

    return _row()


@settings(max_examples=80, deadline=None)
@given(st.lists(row_strategy(), min_size=1, max_size=8))
def test_pack_rows_shift_alignment(rows):
    from rllm_amd.trainer.batch import pack_rows

    batch = pack_rows(rows, device="cpu")
    cu = batch.cu_seqlens
    assert cu[0] == 0 and cu[-1] == batch.input_ids.numel()
    loss_rows = batch.loss_mask.nonzero(as_tuple=True)[0].tolist()
    tgt = batch.targets
    for r in loss_rows:
        # inside some sequence, never its last row
        seq = next(i for i in range(len(rows)) if cu[i] <= r < cu[i + 1])
        assert r + 1 < cu[seq + 1], "loss row at sequence end"
        local = r - cu[seq]
        # target is the NEXT token; that next token is response-masked
        assert int(tgt[r]) == rows[seq].tokens[local + 1]
        assert rows[seq].response_mask[local + 1] == 1
        # advantage/rollout-logprob rows align with the PREDICTED token
        assert abs(float(batch.advantages[r]) - rows[seq].advantages[local + 1]) < 1e-6
        assert abs(float(batch.rollout_logprobs[r])
                   - rows[seq].rollout_logprobs[local + 1]) < 1e-6
    # conservation: every response token whose predecessor exists is a loss row
    expected = sum(1 for row in rows
                   for i, m in enumerate(row.response_mask) if m == 1 and i > 0)
    assert len(loss_rows) == expected


@settings(max_examples=60, deadline=None)
@given(st.lists(row_strategy(), min_size=1, max_size=12),
       st.integers(8, 64))
def test_token_balanced_split_conserves_rows(rows, budget):
    from rllm_amd.trainer.batch import split_rows_token_balanced

    micros = split_rows_token_balanced(rows, budget)
    flat = [r for m in micros for r in m]
    assert sorted(map(id, flat)) == sorted(map(id, rows))  # nothing lost/dup'd
    for m in micros:
        n = sum(len(r.tokens) for r in m)
        assert len(m) == 1 or n <= budget  # only oversized single rows exceed


@settings(max_examples=40, deadline=None)
@given(st.lists(row_strategy(), min_size=1, max_size=6))
def test_pack_rows_positions_restart_per_sequence(rows):
    from rllm_amd.trainer.batch import pack_rows

    batch = pack_rows(rows, device="cpu")
    pos = batch.positions.tolist()
    cu = batch.cu_seqlens
    for i, row in enumerate(rows):
        seg = pos[cu[i] : cu[i + 1]]
        assert seg == list(range(len(row.tokens)))
    assert batch.input_ids.dtype == torch.long
    assert batch.positions.dtype == torch.int32


@given(st.lists(
    st.tuples(st.integers(2, 40),                      # seq length
              st.integers(0, 39)),                     # prompt length (capped below)
    min_size=1, max_size=8))
@settings(max_examples=60, deadline=None)
def test_pack_rows_matches_slow_reference(shapes):
    """The vectorized pack must agree with a per-token reference packer."""
    import numpy as np

    from rllm_amd.trainer.batch import PackedRow, pack_rows

    rng = np.random.default_rng(0)
    rows = []
    for n, p in shapes:
        p = min(p, n - 1)
        toks = rng.integers(0, 1000, n).tolist()
        mask = [0] * p + [1] * (n - p)
        rows.append(PackedRow(tokens=toks, response_mask=mask,
                              advantages=rng.standard_normal(n).tolist(),
                              rollout_logprobs=rng.standard_normal(n).tolist()))

    b = pack_rows(rows)

    # slow reference
    ids, lm, tgt, adv, rl, cu, n_resp = [], [], [], [], [], [0], 0
    for row in rows:
        n = len(row.tokens)
        ids.extend(row.tokens)
        cu.append(cu[-1] + n)
        for r in range(n):
            if r + 1 < n and row.response_mask[r + 1]:
                lm.append(True)
                tgt.append(row.tokens[r + 1])
                adv.append(row.advantages[r + 1])
                rl.append(row.rollout_logprobs[r + 1])
                n_resp += 1
            else:
                lm.append(False)
                tgt.append(0)
                adv.append(0.0)
                rl.append(0.0)

    assert b.input_ids.tolist() == ids
    assert b.cu_seqlens == cu
    assert b.loss_mask.tolist() == lm
    assert b.targets.tolist() == tgt
    assert b.n_response_tokens == n_resp
    import torch

    assert torch.allclose(b.advantages, torch.tensor(adv, dtype=torch.float32))
    assert torch.allclose(b.rollout_logprobs, torch.tensor(rl, dtype=torch.float32))
