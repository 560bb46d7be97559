"""Gateway tests against a fake worker: trace capture, param injection,
field stripping, sessions, routing, weight version, streaming."""

import sys
from pathlib import Path

import httpx
import pytest

sys.path.insert(0, str(Path(__file__).parent))

from helpers.fake_worker import FakeWorkerServer  # noqa: E402

from rllm_amd.gateway.manager import GatewayManager  # noqa: E402
from rllm_amd.gateway.models import GatewayConfig  # noqa: E402
from rllm_amd.gateway.proxy import extract_fields, strip_injected_fields  # noqa: E402
from rllm_amd.gateway.router import StickyLeastLoadedPolicy  # noqa: E402
from rllm_amd.gateway.models import WorkerInfo  # noqa: E402


@pytest.fixture(scope="module")
def stack():
    with FakeWorkerServer() as worker:
        gw = GatewayManager(GatewayConfig())
        gw.start(worker_urls=[worker.url])
        yield gw, worker
        gw.stop()


def test_health_and_workers(stack):
    gw, worker = stack
    r = httpx.get(gw.base_url + "/health")
    assert r.status_code == 200
    r = httpx.get(gw.base_url + "/health/workers")
    assert r.json()["workers"][0]["url"] == worker.url
    assert r.json()["workers"][0]["healthy"] is True


def test_traced_chat_completion(stack):
    gw, worker = stack
    client = gw.client()
    client.create_session("task1:0", sampling_params={"temperature": 0.7})

    url = gw.session_url("task1:0") + "/chat/completions"
    r = httpx.post(url, json={"model": "fake-1.5b", "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 200
    body = r.json()
    # injected fields are stripped from what the client sees
    assert "prompt_token_ids" not in body
    assert "token_ids" not in body["choices"][0]
    assert "logprobs" not in body["choices"][0]  # client didn't ask
    assert body["choices"][0]["message"]["content"] == "fake response"

    traces = client.get_traces("task1:0")
    assert len(traces) == 1
    t = traces[0]
    assert t.completion_token_ids == [7, 8, 9]
    assert t.logprobs == pytest.approx([-0.11, -0.22, -0.33])
    assert len(t.prompt_token_ids) > 0
    assert t.weight_version == 3
    assert t.finish_reason == "stop"

    # the worker saw the injected params + session sampling overrides
    seen = worker.app.state.ctl["requests"][-1]
    assert seen["logprobs"] is True
    assert seen["return_token_ids"] is True
    assert seen["temperature"] == 0.7
    client.delete_session("task1:0")


def test_streaming_trace_assembly(stack):
    gw, worker = stack
    client = gw.client()
    client.create_session("task2:0")
    url = gw.session_url("task2:0") + "/chat/completions"
    with httpx.stream("POST", url, json={"model": "m", "stream": True,
                                         "messages": [{"role": "user", "content": "hi"}]}) as r:
        lines = [ln for ln in r.iter_lines() if ln.startswith("data:")]
    assert lines[-1].strip() == "data: [DONE]"
    traces = client.get_traces("task2:0")
    assert len(traces) == 1
    assert traces[0].completion_token_ids == [7, 8, 9]
    assert traces[0].logprobs == pytest.approx([-0.11, -0.22, -0.33])
    assert traces[0].response_message["content"].startswith("t0 t1")
    client.delete_session("task2:0")


def test_batch_delete_and_weight_version(stack):
    gw, _ = stack
    client = gw.client()
    client.create_session("bd:0")
    client.create_session("bd:1")
    httpx.post(gw.session_url("bd:0") + "/chat/completions",
               json={"model": "m", "messages": []})
    r = httpx.post(gw.base_url + "/sessions/batch_delete", json={"session_ids": ["bd:0", "bd:1"]})
    assert r.status_code == 200
    assert client.get_traces("bd:0") == []

    client.set_weight_version(42)
    r = httpx.get(gw.base_url + "/admin/weight_version")
    assert r.json()["weight_version"] == 42


def test_untracked_plain_proxy(stack):
    gw, _ = stack
    r = httpx.post(gw.base_url + "/v1/chat/completions", json={"model": "m", "messages": []})
    assert r.status_code == 200


def test_extract_fields_openai_style():
    resp = {
        "model": "m",
        "choices": [{
            "message": {"role": "assistant", "content": "x"},
            "finish_reason": "length",
            "logprobs": {"content": [{"token": "a", "logprob": -0.5}, {"token": "b", "logprob": -0.7}]},
            "token_ids": [1, 2],
        }],
        "prompt_token_ids": [9, 8],
    }
    f = extract_fields(resp)
    assert f["logprobs"] == [-0.5, -0.7]
    assert f["completion_token_ids"] == [1, 2]
    assert f["prompt_token_ids"] == [9, 8]
    assert f["finish_reason"] == "length"


def test_strip_preserves_client_logprobs():
    resp = {"choices": [{"logprobs": {"token_logprobs": [-0.1]}, "token_ids": [5]}],
            "prompt_token_ids": [1]}
    out = strip_injected_fields(resp, client_wanted_logprobs=True)
    assert "logprobs" in out["choices"][0]
    assert "token_ids" not in out["choices"][0]
    assert "prompt_token_ids" not in out


def test_sticky_least_loaded_policy():
    policy = StickyLeastLoadedPolicy()
    w1 = WorkerInfo(url="http://a", active_requests=5)
    w2 = WorkerInfo(url="http://b", active_requests=1)
    # new session -> least loaded
    assert policy.pick("s1", [w1, w2]).url == "http://b"
    # sticky on repeat even if load changed
    w2.active_requests = 50
    assert policy.pick("s1", [w1, w2]).url == "http://b"
    # unhealthy sticky worker falls back
    w2.healthy = False
    assert policy.pick("s1", [w1, w2]).url == "http://a"
    # release forgets
    w2.healthy = True
    policy.release_session("s1")
    w2.active_requests = 0
    assert policy.pick("s1", [w1, w2]).url == "http://b"


def test_session_sampling_param_injection_failure_recovery(stack):
    gw, worker = stack
    # injected 500s surface as 500 to the client but don't wedge the gateway
    httpx.post(worker.url + "/admin/fail_next", json={"n": 1})
    r = httpx.post(gw.base_url + "/v1/chat/completions", json={"model": "m", "messages": []})
    assert r.status_code == 500
    r = httpx.post(gw.base_url + "/v1/chat/completions", json={"model": "m", "messages": []})
    assert r.status_code == 200


def test_cumulative_token_mode():
    """Turn>=2 chat requests become pre-tokenized completions calls with the
    prefix-extension invariant."""
    import asyncio

    from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
    from rllm_amd.models.torch_lm import TinyTorchLM
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    model = TinyTorchLM(seed=0)
    handler = make_torch_lm_local_handler(model, parser)
    gw = GatewayManager(GatewayConfig(cumulative_token_mode=True),
                        local_handler=handler, parser=parser)
    gw.start()
    try:
        client = gw.client()
        client.create_session("cum:0", sampling_params={"max_tokens": 5})
        msgs = [{"role": "user", "content": "hi"}]
        r1 = httpx.post(gw.session_url("cum:0") + "/chat/completions",
                        json={"model": "m", "messages": msgs}, timeout=60.0)
        assert r1.status_code == 200
        reply1 = r1.json()["choices"][0]["message"]["content"]
        msgs2 = msgs + [{"role": "assistant", "content": reply1},
                        {"role": "user", "content": "again"}]
        r2 = httpx.post(gw.session_url("cum:0") + "/chat/completions",
                        json={"model": "m", "messages": msgs2}, timeout=60.0)
        assert r2.status_code == 200
        assert r2.json()["choices"][0]["message"]["content"] is not None

        traces = client.get_traces("cum:0")
        assert len(traces) == 2
        t1, t2 = traces
        # prefix-extension: turn 2 prompt == turn 1 prompt + completion + delta
        expected_prefix = t1.prompt_token_ids + t1.completion_token_ids
        assert t2.prompt_token_ids[: len(expected_prefix)] == expected_prefix
        assert len(t2.prompt_token_ids) > len(expected_prefix)  # delta appended
        client.delete_session("cum:0")
    finally:
        gw.stop()


def test_streaming_with_local_handler():
    """stream=True against a colocated local_handler gateway: synthesized SSE
    chunks reach the client AND the trace still captures token ids/logprobs."""
    from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
    from rllm_amd.models.torch_lm import TinyTorchLM
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    handler = make_torch_lm_local_handler(TinyTorchLM(seed=0), parser)
    gw = GatewayManager(GatewayConfig(), local_handler=handler, parser=parser)
    gw.start()
    try:
        client = gw.client()
        client.create_session("st:0", sampling_params={"max_tokens": 4})
        url = gw.session_url("st:0") + "/chat/completions"
        with httpx.stream("POST", url, json={
                "model": "m", "stream": True,
                "messages": [{"role": "user", "content": "hi"}]}, timeout=60.0) as r:
            lines = [ln for ln in r.iter_lines() if ln.startswith("data:")]
        assert lines[-1].strip() == "data: [DONE]"
        assert len(lines) >= 4  # role, content, finish, DONE
        traces = client.get_traces("st:0")
        assert len(traces) == 1
        assert len(traces[0].completion_token_ids) == 4
        assert len(traces[0].logprobs) == 4
        assert traces[0].prompt_token_ids  # injected fields reached the trace
        client.delete_session("st:0")
    finally:
        gw.stop()


def test_cumulative_streaming_local_handler():
    """Cumulative token mode over SSE (colocated path): turn 2 streamed chat
    still becomes a pre-tokenized prefix-extension completions call."""
    from rllm_amd.gateway.native_adapter import make_torch_lm_local_handler
    from rllm_amd.models.torch_lm import TinyTorchLM
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    handler = make_torch_lm_local_handler(TinyTorchLM(seed=0), parser)
    gw = GatewayManager(GatewayConfig(cumulative_token_mode=True),
                        local_handler=handler, parser=parser)
    gw.start()
    try:
        client = gw.client()
        client.create_session("cs:0", sampling_params={"max_tokens": 4})
        url = gw.session_url("cs:0") + "/chat/completions"
        msgs = [{"role": "user", "content": "hi"}]

        def stream_turn(m):
            with httpx.stream("POST", url, json={"model": "m", "stream": True,
                                                 "messages": m}, timeout=60.0) as r:
                lines = [ln for ln in r.iter_lines() if ln.startswith("data:")]
            assert lines[-1].strip() == "data: [DONE]"

        stream_turn(msgs)
        t1 = client.get_traces("cs:0")[0]
        reply1 = t1.response_message["content"]
        msgs2 = msgs + [{"role": "assistant", "content": reply1},
                        {"role": "user", "content": "again"}]
        stream_turn(msgs2)
        traces = client.get_traces("cs:0")
        assert len(traces) == 2
        t1, t2 = traces
        expected_prefix = t1.prompt_token_ids + t1.completion_token_ids
        assert t2.prompt_token_ids[: len(expected_prefix)] == expected_prefix
        assert len(t2.prompt_token_ids) > len(expected_prefix)
        client.delete_session("cs:0")
    finally:
        gw.stop()


def test_token_accumulator_invariant_random_turns():
    """Prefix-extension invariant over randomized multi-turn sessions:
    every continuation's prompt ids start with the previous turn's
    prompt+completion ids, for arbitrary message contents/counts."""
    import random

    from rllm_amd.gateway.token_accumulator import TokenAccumulator
    from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
    from rllm_amd.utils.tokenizer import ByteTokenizer

    parser = QwenChatTemplateParser(ByteTokenizer())
    rng = random.Random(3)
    for trial in range(20):
        acc = TokenAccumulator(parser)
        sid = f"s{trial}"
        msgs = [{"role": "user", "content": f"q{trial}-" + "x" * rng.randint(1, 20)}]
        assert acc.build_prompt_ids(sid, msgs) is None  # first turn: plain path
        prompt_ids = parser.tokenizer.encode(
            "".join(parser.format_message(m) for m in msgs) + parser.assistant_prefix)
        prev = None
        for turn in range(rng.randint(2, 5)):
            completion_ids = [rng.randrange(40, 120) for _ in range(rng.randint(1, 8))]
            acc.record_turn(sid, msgs, prompt_ids, completion_ids)
            expected_prefix = list(prompt_ids) + completion_ids
            reply = "".join(chr(c) for c in completion_ids)
            n_new = rng.randint(1, 2)
            msgs = msgs + [{"role": "assistant", "content": reply}] + [
                {"role": "user", "content": f"t{turn}n{k}-" + "y" * rng.randint(0, 9)}
                for k in range(n_new)]
            nxt = acc.build_prompt_ids(sid, msgs)
            assert nxt is not None
            assert nxt[: len(expected_prefix)] == expected_prefix
            assert len(nxt) > len(expected_prefix)  # the rendered delta
            prev, prompt_ids = expected_prefix, nxt
        # shrunken history is NOT a continuation
        assert acc.build_prompt_ids(sid, msgs[:1]) is None
        acc.reset(sid)
        assert acc.build_prompt_ids(sid, msgs) is None


def test_admin_reload_resets_stickiness(stack):
    gw, worker = stack
    import httpx

    base = gw.base_url
    httpx.post(base + "/sessions", json={"session_id": "sticky-1"})
    httpx.post(base + "/sessions/sticky-1/v1/chat/completions",
               json={"model": "m", "messages": [{"role": "user", "content": "x"}]},
               timeout=30.0)
    r = httpx.post(base + "/admin/reload", timeout=10.0)
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "reloaded" and body["workers"]
    assert all(w["active_sessions"] == 0 for w in body["workers"])
