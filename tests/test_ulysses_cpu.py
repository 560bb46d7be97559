"""Ulysses SP transforms on CPU (gloo, world_size=2): scatter/gather
round-trip, autograd through the all-to-alls, sharded GQA attention ==
dense attention, and kv replication when P > Hk."""

import os

import pytest
import torch
import torch.multiprocessing as mp


def _setup(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = port
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    from rllm_amd.parallel import dist as pdist

    pdist.init_from_env(backend="gloo")
    return pdist


def _causal_gqa_attention(q, k, v, scale):
    """[T, Hq, D] x [T, Hk, D] -> [T, Hq, D], fp32 causal, head-independent."""
    T, Hq, D = q.shape
    Hk = k.shape[1]
    if Hk != Hq:
        rep = Hq // Hk
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    qs, ks, vs = (t.permute(1, 0, 2).float() for t in (q, k, v))
    scores = qs @ ks.transpose(1, 2) * scale
    mask = torch.triu(torch.ones(T, T, dtype=torch.bool), 1)
    scores.masked_fill_(mask, float("-inf"))
    return (scores.softmax(-1) @ vs).permute(1, 0, 2)


def _worker_roundtrip(rank, world, q_out):
    try:
        pdist = _setup(rank, world, "29721")
        from rllm_amd.parallel import ulysses

        T_loc, H, D = 6, 4, 8
        torch.manual_seed(rank)
        x = torch.randn(T_loc, H, D, requires_grad=True)
        y = ulysses.head_to_seq(ulysses.seq_to_head(x, None), None)
        assert torch.allclose(y, x, atol=1e-6)
        # the pair of a2a's is an exact permutation: grad of sum is ones
        y.sum().backward()
        assert torch.allclose(x.grad, torch.ones_like(x))
        pdist.destroy()
        q_out.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q_out.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _worker_attention_parity(rank, world, q_out):
    """Dense fp32 attention on the full batch == ulysses attention on the
    token shard, forward AND backward, incl. the P > Hk replication path."""
    try:
        pdist = _setup(rank, world, "29723")
        from rllm_amd.parallel import ulysses

        T, D, scale = 16, 8, 0.35
        for Hq, Hk in ((4, 2), (4, 1)):  # Hk % P == 0 and P % Hk == 0 paths
            torch.manual_seed(0)  # identical full tensors on both ranks
            qf = torch.randn(T, Hq, D)
            kf = torch.randn(T, Hk, D)
            vf = torch.randn(T, Hk, D)

            ref = _causal_gqa_attention(qf, kf, vf, scale)

            sl = ulysses.shard_slice(T, rank, world)
            q_loc = qf[sl].clone().requires_grad_(True)
            k_loc = kf[sl].clone().requires_grad_(True)
            v_loc = vf[sl].clone().requires_grad_(True)

            out = ulysses.ulysses_attention(
                q_loc, k_loc, v_loc,
                lambda a, b, c: _causal_gqa_attention(a, b, c, scale), None)
            assert torch.allclose(out, ref[sl], atol=1e-5), \
                f"Hq={Hq} Hk={Hk}: {(out - ref[sl]).abs().max()}"

            # backward parity against the dense graph
            qf2 = qf.clone().requires_grad_(True)
            kf2 = kf.clone().requires_grad_(True)
            vf2 = vf.clone().requires_grad_(True)
            loss_w = torch.linspace(0.5, 1.5, T).reshape(T, 1, 1)
            (_causal_gqa_attention(qf2, kf2, vf2, scale) * loss_w).sum().backward()
            (out * loss_w[sl]).sum().backward()
            # NOTE: dense grad wrt q is local (each token's q only affects its
            # own row), so shard grads must match the dense slice exactly.
            assert torch.allclose(q_loc.grad, qf2.grad[sl], atol=1e-5)
            # k/v grads mix across tokens; sum over ranks == dense total
            kv_sum = k_loc.grad.sum() + v_loc.grad.sum()
            total = pdist.all_reduce_scalar(float(kv_sum), op="sum")
            dense_total = float(kf2.grad.sum() + vf2.grad.sum())
            assert abs(total - dense_total) < 1e-3, (total, dense_total)
        pdist.destroy()
        q_out.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q_out.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _run_spawn(fn):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=fn, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(), q.get()]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


@pytest.mark.timeout(180)
def test_a2a_roundtrip_world2():
    _run_spawn(_worker_roundtrip)


@pytest.mark.timeout(180)
def test_ulysses_attention_parity_world2():
    _run_spawn(_worker_attention_parity)


def test_single_rank_passthrough():
    from rllm_amd.parallel import ulysses

    q = torch.randn(8, 4, 6)
    k = torch.randn(8, 2, 6)
    v = torch.randn(8, 2, 6)
    out = ulysses.ulysses_attention(q, k, v, lambda a, b, c: _causal_gqa_attention(a, b, c, 0.4))
    assert torch.allclose(out, _causal_gqa_attention(q, k, v, 0.4))
    with pytest.raises(ValueError):
        ulysses.shard_slice(10, 0, 4)
    # replicate_kv layout
    r = ulysses.replicate_kv(k, 2)
    assert r.shape == (8, 4, 6)
    assert torch.equal(r[:, 0], k[:, 0]) and torch.equal(r[:, 1], k[:, 0])
    assert torch.equal(r[:, 2], k[:, 1])
