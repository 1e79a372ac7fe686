# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/mcp_server.py"]
# ---
# # Stateless MCP server
#
# The MCP shape (reference: 10_integrations/mcp_server_stateless.py, which
# wraps FastMCP): a stateless Model-Context-Protocol server over the
# streamable-HTTP transport — JSON-RPC 2.0 POSTs to one endpoint — exposing
# tools an LLM client can discover (`tools/list`) and invoke (`tools/call`).
# Stateless matters because it maps onto serverless functions: any replica
# can answer any request.  Implemented directly on the web layer (no MCP SDK
# dependency), self-tested over an in-process ASGI transport.

import modal_examples_amd as modal

app = modal.App("example-mcp-server")

TOOLS = [
    {
        "name": "current_date_and_time",
        "description": "Get the current date and time in a timezone.",
        "inputSchema": {
            "type": "object",
            "properties": {"timezone": {"type": "string", "default": "UTC"}},
        },
    },
    {
        "name": "gpu_inventory",
        "description": "Report the GPUs visible to this worker.",
        "inputSchema": {"type": "object", "properties": {}},
    },
]


def _call_tool(name: str, args: dict) -> str:
    if name == "current_date_and_time":
        from datetime import datetime
        from zoneinfo import ZoneInfo

        tz = args.get("timezone", "UTC")
        return datetime.now(ZoneInfo(tz)).isoformat()
    if name == "gpu_inventory":
        import torch

        n = torch.cuda.device_count() if torch.cuda.is_available() else 0
        names = [torch.cuda.get_device_name(i) for i in range(n)]
        return f"{n} GPU(s): {names}" if n else "no GPUs visible"
    raise ValueError(f"unknown tool {name!r}")


@app.function()
@modal.asgi_app(label="mcp")
def mcp_app():
    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse

    api = FastAPI()

    @api.post("/mcp")
    async def mcp(request: Request):
        req = await request.json()
        rid, method = req.get("id"), req.get("method")
        params = req.get("params") or {}

        def ok(result):
            return JSONResponse({"jsonrpc": "2.0", "id": rid, "result": result})

        def err(code, message):
            return JSONResponse({"jsonrpc": "2.0", "id": rid,
                                 "error": {"code": code, "message": message}})

        if method == "initialize":
            return ok({"protocolVersion": params.get("protocolVersion", "2025-03-26"),
                       "capabilities": {"tools": {}},
                       "serverInfo": {"name": "mi355x-mcp", "version": "1.0"}})
        if method == "notifications/initialized":
            return JSONResponse({}, status_code=202)
        if method == "tools/list":
            return ok({"tools": TOOLS})
        if method == "tools/call":
            try:
                text = _call_tool(params.get("name", ""),
                                  params.get("arguments") or {})
                return ok({"content": [{"type": "text", "text": text}],
                           "isError": False})
            except Exception as e:  # tool errors are results, not RPC errors
                return ok({"content": [{"type": "text", "text": str(e)}],
                           "isError": True})
        return err(-32601, f"method not found: {method}")

    return api


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    asgi = mcp_app.raw()

    def rpc(client, method, params=None, rid=1):
        return client.post("/mcp", json={"jsonrpc": "2.0", "id": rid,
                                         "method": method,
                                         "params": params or {}})

    async def go():
        transport = httpx.ASGITransport(app=asgi)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://mcp") as c:
            init = (await rpc(c, "initialize")).json()["result"]
            assert init["serverInfo"]["name"] == "mi355x-mcp"
            tools = (await rpc(c, "tools/list")).json()["result"]["tools"]
            assert {t["name"] for t in tools} == {"current_date_and_time",
                                                  "gpu_inventory"}
            r = (await rpc(c, "tools/call",
                           {"name": "current_date_and_time",
                            "arguments": {"timezone": "UTC"}})).json()["result"]
            assert not r["isError"] and "T" in r["content"][0]["text"]
            g = (await rpc(c, "tools/call",
                           {"name": "gpu_inventory"})).json()["result"]
            bad = (await rpc(c, "no/such")).json()
            assert bad["error"]["code"] == -32601
            return r["content"][0]["text"], g["content"][0]["text"]

    now, gpus = asyncio.run(go())
    print(f"time tool → {now}")
    print(f"gpu tool  → {gpus}")
    print("MCP server OK (initialize / tools/list / tools/call)")
