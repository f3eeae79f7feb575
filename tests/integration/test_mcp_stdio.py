"""SDK-free stdio MCP client against a scripted JSON-RPC server subprocess
(the MCP stdio wire protocol spoken for real), plus the full mesh path:
an MCPToolbox node whose live session is DIALED by the worker, its tools
namespaced and invoked by an agent over the wire."""

import sys
import textwrap

import pytest

SERVER = textwrap.dedent("""
    import json, sys

    def send(obj):
        sys.stdout.write(json.dumps(obj) + "\\n")
        sys.stdout.flush()

    for line in sys.stdin:
        line = line.strip()
        if not line:
            continue
        msg = json.loads(line)
        m, mid = msg.get("method"), msg.get("id")
        if m == "initialize":
            send({"jsonrpc": "2.0", "id": mid, "result": {
                "protocolVersion": msg["params"]["protocolVersion"],
                "capabilities": {"tools": {}},
                "serverInfo": {"name": "scripted-mcp", "version": "1.0"}}})
        elif m == "notifications/initialized":
            pass
        elif m == "tools/list":
            send({"jsonrpc": "2.0", "id": mid, "result": {"tools": [
                {"name": "add", "description": "Add two integers.",
                 "inputSchema": {"type": "object", "properties": {
                     "a": {"type": "integer"}, "b": {"type": "integer"}},
                     "required": ["a", "b"]}}]}})
        elif m == "tools/call":
            p = msg["params"]
            if p["name"] != "add":
                send({"jsonrpc": "2.0", "id": mid,
                      "error": {"code": -32601, "message": "no such tool"}})
            else:
                a, b = p["arguments"]["a"], p["arguments"]["b"]
                send({"jsonrpc": "2.0", "id": mid, "result": {
                    "content": [{"type": "text", "text": str(a + b)}],
                    "isError": False}})
        else:
            send({"jsonrpc": "2.0", "id": mid,
                  "error": {"code": -32601, "message": f"unknown {m}"}})
""")


@pytest.fixture()
def server_script(tmp_path):
    path = tmp_path / "mcp_server.py"
    path.write_text(SERVER)
    return str(path)


async def test_stdio_session_speaks_the_protocol(server_script):
    from calfkit_amd.adapters.mcp import StdioMCPSession

    s = await StdioMCPSession.open(sys.executable, (server_script,))
    try:
        assert s.server_info["name"] == "scripted-mcp"
        tools = await s.list_tools()
        assert [t.name for t in tools] == ["add"]
        assert tools[0].input_schema["required"] == ["a", "b"]
        r = await s.call_tool("add", {"a": 19, "b": 23})
        assert r.content == "42" and not r.is_error
        with pytest.raises(RuntimeError):
            await s.call_tool("nope", {})
    finally:
        await s.close()


async def test_toolbox_node_dials_stdio_server_over_the_mesh(server_script):
    """End-to-end: Worker hosts the MCPToolbox (dialing the live stdio
    server itself via open_session), an agent invokes the namespaced tool
    through the wire, and the subprocess computes the reply."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import (
        InMemoryBus,
        InMemoryDirectory,
        InProcessThreadStore,
    )
    from calfkit_amd.definitions.selections import Toolboxes
    from calfkit_amd.definitions.toolbox import MCPToolbox
    from calfkit_amd.testing import FakeProvider, call_tool, reply

    box = MCPToolbox(name="math", transport="stdio",
                     command=sys.executable, args=(server_script,))
    provider = FakeProvider([call_tool("math.add", a=40, b=2),
                             reply("the answer is 42")])
    agent = Agent(name="asker", model_client=provider,
                  tools=Toolboxes(names=("math",)))
    bus, store, d = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    async with Worker(agent, box, bus=bus, store=store, directory=d):
        client = await Client.connect(bus, d)
        result = await client.execute("asker", "add 40 and 2", timeout=60)
        assert result.text == "the answer is 42"
        # the tool round really went through the subprocess
        req = provider.requests[-1]
        tool_msgs = [m for m in req.messages if m.role == "tool"]
        assert tool_msgs and "42" in str(tool_msgs[-1].parts[0])


async def test_sse_session_speaks_the_protocol():
    """Native SSE MCP dialing: endpoint event -> JSON-RPC POSTs ->
    message events on the live stream (no SDK; a real uvicorn server on
    a loopback port, because buffered test transports cannot stream)."""
    import asyncio
    import json as _json

    import pytest as _pytest
    uvicorn = _pytest.importorskip("uvicorn")
    from calfkit_amd.adapters.mcp import SseMCPSession

    queue: asyncio.Queue = asyncio.Queue()

    def sse(event, obj):
        payload = _json.dumps(obj) if not isinstance(obj, str) else obj
        return f"event: {event}\ndata: {payload}\n\n".encode()

    async def app(scope, receive, send):
        if scope["type"] != "http":
            return
        if scope["method"] == "GET":
            await send({"type": "http.response.start", "status": 200,
                        "headers": [(b"content-type", b"text/event-stream")]})
            await send({"type": "http.response.body",
                        "body": sse("endpoint", "/messages"),
                        "more_body": True})
            while True:
                item = await queue.get()
                if item is None:
                    break
                await send({"type": "http.response.body",
                            "body": sse("message", item),
                            "more_body": True})
            await send({"type": "http.response.body", "body": b""})
            return
        body = b""
        while True:
            ev = await receive()
            body += ev.get("body", b"")
            if not ev.get("more_body"):
                break
        msg = _json.loads(body)
        m, mid = msg.get("method"), msg.get("id")
        if m == "initialize":
            await queue.put({"jsonrpc": "2.0", "id": mid, "result": {
                "protocolVersion": msg["params"]["protocolVersion"],
                "capabilities": {"tools": {}},
                "serverInfo": {"name": "sse-mcp", "version": "1.0"}}})
        elif m == "tools/list":
            await queue.put({"jsonrpc": "2.0", "id": mid, "result": {
                "tools": [{"name": "mul", "description": "Multiply.",
                           "inputSchema": {"type": "object"}}]}})
        elif m == "tools/call":
            a = msg["params"]["arguments"]
            await queue.put({"jsonrpc": "2.0", "id": mid, "result": {
                "content": [{"type": "text",
                             "text": str(a["a"] * a["b"])}]}})
        await send({"type": "http.response.start", "status": 202,
                    "headers": []})
        await send({"type": "http.response.body", "body": b""})

    config = uvicorn.Config(app, host="127.0.0.1", port=0, log_level="error",
                            lifespan="off")
    server = uvicorn.Server(config)
    task = asyncio.ensure_future(server.serve())
    while not server.started:
        await asyncio.sleep(0.02)
    port = server.servers[0].sockets[0].getsockname()[1]

    s = await SseMCPSession.open(f"http://127.0.0.1:{port}/sse")
    try:
        assert s.server_info["name"] == "sse-mcp"
        tools = await s.list_tools()
        assert [t.name for t in tools] == ["mul"]
        r = await s.call_tool("mul", {"a": 6, "b": 7})
        assert r.content == "42" and not r.is_error
    finally:
        await queue.put(None)
        await s.close()
        server.should_exit = True
        await asyncio.wait_for(task, 5)


async def test_stdio_session_passes_toolbox_env(tmp_path):
    """mcp.json `env` entries reach the stdio server subprocess."""
    import sys as _sys

    from calfkit_amd.adapters.mcp import StdioMCPSession

    script = tmp_path / "envserver.py"
    script.write_text(
        "import json, os, sys\n"
        "for line in sys.stdin:\n"
        "    msg = json.loads(line)\n"
        "    if msg.get('method') == 'initialize':\n"
        "        sys.stdout.write(json.dumps({'jsonrpc': '2.0',\n"
        "            'id': msg['id'], 'result': {'serverInfo':\n"
        "            {'name': os.environ.get('CALF_MCP_TOKEN', '?')}}})\n"
        "            + '\\n')\n"
        "        sys.stdout.flush()\n")
    s = await StdioMCPSession.open(_sys.executable, (str(script),),
                                  env={"CALF_MCP_TOKEN": "sesame"})
    try:
        assert s.server_info["name"] == "sesame"
    finally:
        await s.close()
