"""Minimal MCP stdio server for tests: newline-delimited JSON-RPC with an
`add` tool and an `explode` tool that reports isError."""

import json
import sys


def main():
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        msg = json.loads(line)
        method = msg.get("method")
        if "id" not in msg:  # notification
            continue
        if method == "initialize":
            result = {"protocolVersion": msg["params"]["protocolVersion"],
                      "capabilities": {"tools": {}},
                      "serverInfo": {"name": "fake-mcp", "version": "0.0"}}
        elif method == "tools/list":
            result = {"tools": [
                {"name": "add", "description": "Add two numbers",
                 "inputSchema": {"type": "object",
                                 "properties": {"a": {"type": "number"},
                                                "b": {"type": "number"}},
                                 "required": ["a", "b"]}},
                {"name": "explode", "description": "Always fails",
                 "inputSchema": {"type": "object", "properties": {}}},
            ]}
        elif method == "tools/call":
            name = msg["params"]["name"]
            args = msg["params"].get("arguments", {})
            if name == "add":
                result = {"content": [{"type": "text",
                                       "text": str(args["a"] + args["b"])}]}
            elif name == "explode":
                result = {"content": [{"type": "text", "text": "boom"}],
                          "isError": True}
            else:
                print(json.dumps({"jsonrpc": "2.0", "id": msg["id"],
                                  "error": {"code": -32601, "message": f"no tool {name}"}}),
                      flush=True)
                continue
        else:
            print(json.dumps({"jsonrpc": "2.0", "id": msg["id"],
                              "error": {"code": -32601, "message": f"no method {method}"}}),
                  flush=True)
            continue
        print(json.dumps({"jsonrpc": "2.0", "id": msg["id"], "result": result}), flush=True)


if __name__ == "__main__":
    main()
