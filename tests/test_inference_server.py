"""OpenAI-compatible inference server over a stubbed engine driver (CPU):
wire format, TITO completions, prompt-length guard, weight version."""

import sys
from pathlib import Path

import httpx
import pytest

sys.path.insert(0, str(Path(__file__).parent))

from rllm_amd.engine.inference.llm_engine import RequestOutput, SamplingParams
from rllm_amd.engine.inference.server import InferenceServer
from rllm_amd.parser.chat_template_parser import QwenChatTemplateParser
from rllm_amd.utils.tokenizer import ByteTokenizer


class StubDriver:
    """Duck-typed AsyncEngineDriver returning canned token streams."""

    class _Engine:
        weight_version = 7

    engine = _Engine()

    async def submit(self, prompt_ids, params: SamplingParams, request_id=None):
        n = min(3, params.max_tokens)
        return RequestOutput(request_id=request_id or "r", prompt_ids=list(prompt_ids),
                             token_ids=[104, 105, 33][:n], logprobs=[-0.3, -0.2, -0.1][:n],
                             finish_reason="length", weight_version=7)

    def shutdown(self):
        pass


@pytest.fixture(scope="module")
def server():
    parser = QwenChatTemplateParser(ByteTokenizer())
    srv = InferenceServer.__new__(InferenceServer)
    # build without creating a real AsyncEngineDriver thread
    import socket

    from rllm_amd.engine.inference.server import create_server_app

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        srv.host, srv.port = "127.0.0.1", s.getsockname()[1]
    srv.driver = StubDriver()
    srv.app = create_server_app(srv.driver, parser, "stub-model", max_prompt_length=64)
    srv._server = None
    srv._thread = None
    srv.start()
    yield srv
    srv.stop()


def test_health(server):
    r = httpx.get(server.url + "/health")
    assert r.json()["model"] == "stub-model"
    assert r.json()["weight_version"] == 7


def test_chat_completions_wire_format(server):
    r = httpx.post(server.url + "/v1/chat/completions",
                   json={"model": "m", "messages": [{"role": "user", "content": "hi"}],
                         "max_tokens": 3})
    body = r.json()
    assert r.status_code == 200
    ch = body["choices"][0]
    assert ch["token_ids"] == [104, 105, 33]
    assert ch["logprobs"]["token_logprobs"] == [-0.3, -0.2, -0.1]
    assert ch["finish_reason"] == "length"
    assert body["prompt_token_ids"]
    assert body["weight_version"] == 7
    assert ch["message"]["content"] == "hi!"  # bytes 104,105,33 decode to 'hi!'
    assert body["usage"]["completion_tokens"] == 3


def test_completions_tito(server):
    r = httpx.post(server.url + "/v1/completions",
                   json={"model": "m", "prompt_token_ids": [1, 2, 3], "max_tokens": 2})
    body = r.json()
    assert body["prompt_token_ids"] == [1, 2, 3]
    assert body["choices"][0]["token_ids"] == [104, 105]


def test_max_prompt_length_guard(server):
    r = httpx.post(server.url + "/v1/chat/completions",
                   json={"model": "m",
                         "messages": [{"role": "user", "content": "x" * 500}]})
    assert r.status_code == 400
    assert "MAX_PROMPT_LENGTH_EXCEEDED" in r.text
