"""A minimal stdio MCP server for tests: newline-delimited JSON-RPC,
initialize/tools-list/tools-call, two tools (one read-only, one
destructive-named), plus a stray non-JSON log line to exercise the
client's tolerance. Run as: python tests/fake_mcp_server.py"""
import json
import os
import sys

TOOLS = [
    {"name": "get_weather",
     "description": "Read-only: report the weather for a city.",
     "inputSchema": {"type": "object",
                     "properties": {"city": {"type": "string",
                                             "description": "city name"}}}},
    {"name": "delete_everything",
     "description": "Destructive: deletes it all.",
     "inputSchema": {"type": "object", "properties": {}}},
]


def reply(obj):
    sys.stdout.write(json.dumps(obj) + "\n")
    sys.stdout.flush()


def main():
    sys.stdout.write("fake-mcp booting (stray log line)\n")
    sys.stdout.flush()
    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        msg = json.loads(line)
        method = msg.get("method")
        rid = msg.get("id")
        if method == "initialize":
            reply({"jsonrpc": "2.0", "id": rid, "result": {
                "protocolVersion": msg["params"]["protocolVersion"],
                "capabilities": {"tools": {}},
                "serverInfo": {"name": "fake-mcp", "version": "1.0"}}})
        elif method == "notifications/initialized":
            continue
        elif method == "tools/list":
            reply({"jsonrpc": "2.0", "id": rid, "result": {"tools": TOOLS}})
        elif method == "tools/call":
            name = msg["params"]["name"]
            args = msg["params"].get("arguments", {})
            if name == "get_weather":
                token = os.environ.get("FAKE_MCP_TOKEN", "")
                reply({"jsonrpc": "2.0", "id": rid, "result": {
                    "content": [{"type": "text",
                                 "text": f"sunny in {args.get('city', '?')}"
                                         + (f" [token={token}]" if token else "")}]}})
            elif name == "delete_everything":
                reply({"jsonrpc": "2.0", "id": rid, "result": {
                    "content": [{"type": "text", "text": "deleted"}]}})
            else:
                reply({"jsonrpc": "2.0", "id": rid, "result": {
                    "isError": True,
                    "content": [{"type": "text", "text": f"no tool {name}"}]}})
        else:
            reply({"jsonrpc": "2.0", "id": rid,
                   "error": {"code": -32601, "message": f"unknown {method}"}})


if __name__ == "__main__":
    main()
