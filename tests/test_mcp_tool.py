"""MCP stdio client + tool adapter against a fake in-repo MCP server."""

import sys
from pathlib import Path

import pytest

from rllm_amd.tools.mcp_tool import MCPClient, MCPError, connect_mcp_tools

SERVER = [sys.executable, str(Path(__file__).parent / "helpers" / "fake_mcp_server.py")]


def test_handshake_and_list():
    with MCPClient(SERVER) as c:
        assert c.server_info["serverInfo"]["name"] == "fake-mcp"
        tools = c.list_tools()
        assert [t["name"] for t in tools] == ["add", "explode"]


def test_tool_adapter_call_and_schema():
    client, tools = connect_mcp_tools(SERVER)
    try:
        add = {t.name: t for t in tools}["add"]
        assert add.json["function"]["parameters"]["required"] == ["a", "b"]
        out = add.forward(a=2, b=40)
        assert out.error is None
        assert out.output == "42"
        # repeated calls over the same pipe
        assert add.forward(a=1, b=1).output == "2"
    finally:
        client.close()


def test_is_error_and_unknown():
    client, tools = connect_mcp_tools(SERVER)
    try:
        boom = {t.name: t for t in tools}["explode"]
        out = boom.forward()
        assert out.error == "boom"
        with pytest.raises(MCPError):
            client.call_tool("nope", {})
    finally:
        client.close()


def test_server_death_raises():
    client, tools = connect_mcp_tools(SERVER)
    client._proc.kill()
    client._proc.wait()
    out = tools[0].forward(a=1, b=2)
    assert out.error is not None
