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


def test_provider_proxy_routing_and_translation():
    """Provider proxy (reference _litellm_server role): model-prefix
    routing, key injection, Anthropic request/response translation."""
    from rllm_amd.eval.provider_proxy import ProviderProxy

    seen = []

    def fake(url, payload, headers):
        seen.append((url, payload, headers))
        if "/v1/messages" in url:
            return 200, {"id": "m1", "content": [{"type": "text", "text": "claude says hi"}],
                         "stop_reason": "end_turn",
                         "usage": {"input_tokens": 7, "output_tokens": 3}}
        return 200, {"id": "c1", "object": "chat.completion", "model": payload["model"],
                     "choices": [{"index": 0, "message": {"role": "assistant", "content": "ok"},
                                  "finish_reason": "stop"}]}

    proxy = ProviderProxy({
        "openai": {"base_url": "https://api.openai.example/v1", "api_key": "sk-o"},
        "anthropic": {"base_url": "https://api.anthropic.example",
                      "api_key": "sk-a", "style": "anthropic"},
    }, transport=fake)

    # openai passthrough: prefix stripped, bearer injected
    status, resp = proxy.handle_chat({"model": "openai/gpt-x",
                                      "messages": [{"role": "user", "content": "hi"}]})
    assert status == 200 and resp["choices"][0]["message"]["content"] == "ok"
    url, payload, headers = seen[-1]
    assert url.endswith("/chat/completions") and payload["model"] == "gpt-x"
    assert headers["Authorization"] == "Bearer sk-o"

    # anthropic: translated both ways, system lifted out of messages
    status, resp = proxy.handle_chat({
        "model": "anthropic/claude-x", "max_tokens": 32, "temperature": 0.5,
        "messages": [{"role": "system", "content": "be brief"},
                     {"role": "user", "content": "hi"}]})
    url, payload, headers = seen[-1]
    assert url.endswith("/v1/messages") and headers["x-api-key"] == "sk-a"
    assert payload["system"] == "be brief" and payload["temperature"] == 0.5
    assert all(m["role"] != "system" for m in payload["messages"])
    assert status == 200
    assert resp["object"] == "chat.completion"
    assert resp["choices"][0]["message"]["content"] == "claude says hi"
    assert resp["choices"][0]["finish_reason"] == "stop"
    assert resp["usage"]["prompt_tokens"] == 7

    # unprefixed model falls to the default provider
    proxy.handle_chat({"model": "bare-model", "messages": []})
    assert seen[-1][1]["model"] == "bare-model"


def test_provider_proxy_http_server():
    """The proxy serves a real OpenAI-compatible endpoint."""
    import httpx

    from rllm_amd.eval.provider_proxy import ProviderProxy

    def fake(url, payload, headers):
        return 200, {"id": "c", "object": "chat.completion", "model": payload["model"],
                     "choices": [{"index": 0, "message": {"role": "assistant", "content": "pong"},
                                  "finish_reason": "stop"}]}

    proxy = ProviderProxy({"local": {"base_url": "http://up/v1"}}, transport=fake)
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base = proxy.start(port=port)
    try:
        import time

        for _ in range(50):
            try:
                r = httpx.get(base.replace("/v1", "/health"), timeout=1.0)
                break
            except Exception:  # noqa: BLE001
                time.sleep(0.1)
        r = httpx.post(base + "/chat/completions",
                       json={"model": "m", "messages": [{"role": "user", "content": "ping"}]},
                       timeout=10.0)
        assert r.status_code == 200
        assert r.json()["choices"][0]["message"]["content"] == "pong"
    finally:
        proxy.stop()
