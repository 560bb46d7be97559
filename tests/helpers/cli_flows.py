"""Flows/evaluators referenced by CLI tests via module:attr resolution."""

import httpx

import rllm_amd


@rllm_amd.rollout
def flow(task, config):
    r = httpx.post(config.base_url + "/chat/completions",
                   json={"model": config.model,
                         "messages": [{"role": "user", "content": str(task.instruction)}]},
                   timeout=30.0)
    r.raise_for_status()
    return None


@rllm_amd.evaluator
def ev(task, episode):
    step = episode.trajectories[0].steps[-1]
    if not step.response_ids:
        return 0.0
    return float(sum(1 for t in step.response_ids if t % 2 == 0) / len(step.response_ids))
