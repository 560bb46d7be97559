"""Advantage estimators, transform, rejection sampling."""

import numpy as np
import pytest

from rllm_amd.trainer.algorithms.advantage import (
    collect_reward_and_advantage_from_trajectory_groups,
    get_rllm_adv_estimator,
)
from rllm_amd.trainer.algorithms.config import (
    AlgorithmConfig,
    CompactFilteringConfig,
    RejectionSamplingConfig,
    TransformConfig,
    rLLMAdvantageEstimator,
)
from rllm_amd.trainer.algorithms.rejection_sampling import (
    RejectionSamplingState,
    apply_rejection_sampling_and_filtering,
)
from rllm_amd.trainer.algorithms.rl_algo import (
    calculate_grpo_advantages_per_group,
    calculate_rloo_advantages_per_group,
)
from rllm_amd.trainer.algorithms.transform import transform_episodes_to_trajectory_groups
from rllm_amd.types import Episode, Step, Trajectory
from rllm_amd.workflows.workflow import TerminationReason


def test_grpo_formula():
    r = np.array([1.0, 0.0, 1.0, 0.0])
    adv, ret = calculate_grpo_advantages_per_group(r)
    expected = (r - 0.5) / (0.5 + 1e-6)
    np.testing.assert_allclose(adv, expected, rtol=1e-6)


def test_grpo_no_norm():
    r = np.array([1.0, 0.0])
    adv, _ = calculate_grpo_advantages_per_group(r, norm_adv_by_std_in_grpo=False)
    np.testing.assert_allclose(adv, [0.5, -0.5])


def test_grpo_single_trajectory():
    r = np.array([0.7])
    adv, _ = calculate_grpo_advantages_per_group(r)
    np.testing.assert_allclose(adv, [0.7 / (1.0 + 1e-6)], rtol=1e-6)


def test_rloo_formula():
    r = np.array([1.0, 0.0, 0.0, 0.0])
    adv, _ = calculate_rloo_advantages_per_group(r)
    # n/(n-1) * (r - mean) == r - LOO-mean
    expected = 4 / 3 * (r - 0.25)
    np.testing.assert_allclose(adv, expected)


def make_episodes(task_rewards: dict[str, list[float]], name="solver"):
    """task_rewards: task_id -> list of per-rollout rewards."""
    episodes = []
    for task_id, rewards in task_rewards.items():
        for idx, rew in enumerate(rewards):
            step = Step(
                prompt_ids=[1, 2],
                response_ids=[3, 4, 5],
                logprobs=[-0.1] * 3,
                chat_completions=[{"role": "user", "content": "q"}],
                reward=rew,
                done=True,
            )
            traj = Trajectory(name=name, steps=[step], reward=rew)
            episodes.append(
                Episode(
                    id=f"{task_id}:{idx}",
                    task={"q": task_id},
                    trajectories=[traj],
                    is_correct=rew > 0,
                    termination_reason=TerminationReason.ENV_DONE,
                )
            )
    return episodes


def test_transform_groups_by_task_and_name():
    eps = make_episodes({"a": [1.0, 0.0], "b": [0.0, 0.0]})
    groups, metrics = transform_episodes_to_trajectory_groups(eps, TransformConfig())
    assert len(groups) == 2
    ids = sorted(g.group_id for g in groups)
    assert ids == ["a:solver", "b:solver"]
    assert metrics["groups/num_groups"] == 2
    assert metrics["groups/num_trajs_after_filter"] == 4


def test_transform_compact_filtering_masks_episodes():
    eps = make_episodes({"a": [1.0, 0.0]})
    for ep in eps:
        ep.termination_reason = TerminationReason.MAX_RESPONSE_LENGTH_EXCEEDED
    cf = CompactFilteringConfig(enable=True, mask_max_response_length_exceeded=True)
    groups, _ = transform_episodes_to_trajectory_groups(eps, TransformConfig(), cf)
    assert groups == []


def test_transform_propagates_reward_from_last_step():
    eps = make_episodes({"a": [1.0, 0.0]})
    for ep in eps:
        ep.trajectories[0].reward = None
    groups, _ = transform_episodes_to_trajectory_groups(eps, TransformConfig())
    rewards = sorted(t.reward for g in groups for t in g.trajectories)
    assert rewards == [0.0, 1.0]


def test_collect_advantage_writes_steps():
    eps = make_episodes({"a": [1.0, 0.0, 1.0, 0.0]})
    groups, _ = transform_episodes_to_trajectory_groups(eps, TransformConfig())
    cfg = AlgorithmConfig(estimator=rLLMAdvantageEstimator.GRPO)
    metrics = collect_reward_and_advantage_from_trajectory_groups(groups, cfg)
    advs = [t.steps[0].advantage for g in groups for t in g.trajectories]
    assert all(isinstance(a, float) for a in advs)
    expected = (np.array([1.0, 0.0, 1.0, 0.0]) - 0.5) / (0.5 + 1e-6)
    np.testing.assert_allclose(sorted(advs), sorted(expected), rtol=1e-5)
    assert metrics["reward/solver/mean"] == 0.5
    # advantage write-back reaches the source episodes (reference passes trajs by reference)
    assert eps[0].trajectories[0].steps[0].advantage is not None


def test_estimator_registry_unknown_raises():
    with pytest.raises(ValueError):
        get_rllm_adv_estimator("nope")


def test_rejection_sampling_none_mode():
    eps = make_episodes({"a": [1.0, 0.0], "b": [1.0, 1.0]})
    groups, _ = transform_episodes_to_trajectory_groups(eps, TransformConfig())
    state = RejectionSamplingState()
    fg, fe, metrics = apply_rejection_sampling_and_filtering(eps, groups, RejectionSamplingConfig(mode="none"), state)
    assert len(fg) == 2
    assert metrics["batch/solve_partial"] == 0.5


def test_rejection_sampling_episode_mode_accumulates():
    # First batch: no partial solves -> returns empty, accumulates
    eps1 = make_episodes({"b": [1.0, 1.0]})
    groups1, _ = transform_episodes_to_trajectory_groups(eps1, TransformConfig())
    state = RejectionSamplingState()
    cfg = RejectionSamplingConfig(mode="episode", min_partial_solve_tasks=1)
    fg, fe, _ = apply_rejection_sampling_and_filtering(eps1, groups1, cfg, state)
    assert fg == [] and fe == []
    # Second batch has a partial solve -> returns accumulated groups
    eps2 = make_episodes({"a": [1.0, 0.0]})
    groups2, _ = transform_episodes_to_trajectory_groups(eps2, TransformConfig())
    fg, fe, _ = apply_rejection_sampling_and_filtering(eps2, groups2, cfg, state)
    assert len(fg) == 2  # both accumulated groups


def test_rejection_sampling_min_trajs():
    eps = make_episodes({"a": [1.0]})  # single-traj group
    groups, _ = transform_episodes_to_trajectory_groups(eps, TransformConfig())
    state = RejectionSamplingState()
    cfg = RejectionSamplingConfig(mode="none", min_trajs_per_group=2)
    fg, fe, metrics = apply_rejection_sampling_and_filtering(eps, groups, cfg, state)
    assert fg == []
    assert metrics["batch/groups_dropped_insufficient_trajs"] == 1


def test_tis_weights_token_and_sequence():
    import torch

    from rllm_amd.trainer.policy import tis_weights

    old = torch.tensor([0.0, -1.0, -2.0, -0.5])
    roll = torch.tensor([-0.5, -1.0, -1.0, -0.5])
    # token mode: exp(old-roll) clamped
    w = tis_weights(old, roll, mode="token", cap=1.2)
    expect = torch.exp(old - roll).clamp(max=1.2)
    assert torch.allclose(w, expect)
    # sequence mode: rows 0-1 = seq 0, rows 2-3 = seq 1
    seq_ids = torch.tensor([0, 0, 1, 1])
    w = tis_weights(old, roll, mode="sequence", cap=10.0, seq_ids=seq_ids)
    w0 = torch.exp((old - roll)[:2].sum()).clamp(max=10.0)
    w1 = torch.exp((old - roll)[2:].sum()).clamp(max=10.0)
    assert torch.allclose(w, torch.stack([w0, w0, w1, w1]))
    # disabled / errors
    assert tis_weights(old, roll, mode=None, cap=1.0) is None
    import pytest

    with pytest.raises(ValueError):
        tis_weights(old, roll, mode="sequence", cap=1.0)
    with pytest.raises(ValueError):
        tis_weights(old, roll, mode="banana", cap=1.0)


def test_advantage_properties_random():
    """Estimator invariants over random groups: GRPO advantages are
    mean-zero per group; RLOO baselines are leave-one-out means;
    uniform-reward groups give (near-)zero advantage."""
    import random

    rng = random.Random(0)
    for trial in range(50):
        n = rng.randint(2, 9)
        r = np.array([rng.choice([0.0, 0.5, 1.0]) for _ in range(n)])
        grpo, _ = calculate_grpo_advantages_per_group(r)
        assert abs(float(np.sum(grpo))) < 1e-4 * n
        rloo, _ = calculate_rloo_advantages_per_group(r)
        for i in range(n):
            others = np.delete(r, i)
            assert abs(float(rloo[i]) - (r[i] - float(others.mean()))) < 1e-6
        uniform = np.full(n, 0.7)
        g, _ = calculate_grpo_advantages_per_group(uniform)
        l, _ = calculate_rloo_advantages_per_group(uniform)
        assert np.all(np.abs(g) < 1e-4) and np.all(np.abs(l) < 1e-6)


def test_group_mode_rejection_sampling_accumulates():
    """Group mode (a NotImplementedError stub in the reference): keep only
    informative groups, accumulate across batches, release at
    min_groups_per_batch."""
    from rllm_amd.trainer.algorithms.config import RejectionSamplingConfig
    from rllm_amd.trainer.algorithms.rejection_sampling import (
        RejectionSamplingState,
        apply_rejection_sampling_and_filtering,
    )
    from rllm_amd.types import Episode, Step, Trajectory, TrajectoryGroup

    def make(task_id, rewards):
        eps, trajs = [], []
        for i, r in enumerate(rewards):
            st = Step(prompt_ids=[1, 2], response_ids=[3, 4], logprobs=[-0.1, -0.1],
                      reward=r, done=True)
            tr = Trajectory(name="s", steps=[st], reward=r)
            trajs.append(tr)
            eps.append(Episode(id=f"{task_id}:{i}", trajectories=[tr], is_correct=r > 0))
        return eps, TrajectoryGroup(group_id=f"{task_id}:s", trajectories=trajs)

    cfg = RejectionSamplingConfig(mode="group", filter_uniform_groups=True,
                                  min_groups_per_batch=2)
    state = RejectionSamplingState()

    # batch 1: one informative group + one uniform (dropped) -> accumulate
    e1, g1 = make("a", [0.0, 1.0])
    e2, g2 = make("b", [1.0, 1.0])
    groups, eps, m = apply_rejection_sampling_and_filtering(e1 + e2, [g1, g2], cfg, state)
    assert groups == [] and len(state.accumulated_groups) == 1

    # batch 2: another informative group -> release both
    e3, g3 = make("c", [0.0, 1.0])
    groups, eps, m = apply_rejection_sampling_and_filtering(e3, [g3], cfg, state)
    assert [g.group_id for g in groups] == ["a:s", "c:s"]
    assert m["batch/groups_after_filter"] == 2
