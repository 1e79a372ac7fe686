"""Web layer: ingress routes for fastapi/asgi/wsgi endpoints, class-based
endpoints, and the OpenAI app (hermetic, via httpx ASGI transport)."""
import asyncio

import httpx
import pytest

import modal_examples_amd as modal
from modal_examples_amd.web.ingress import build_ingress_app

app = modal.App("test-web")


@app.function()
@modal.fastapi_endpoint(method="GET", label="double")
def double(x: int = 1):
    return {"doubled": x * 2}


@app.function()
@modal.fastapi_endpoint(method="POST", label="concat")
def concat(a: str = "x", b: str = "y"):
    return {"joined": a + b}


@app.function()
@modal.asgi_app(label="sub")
def sub_app():
    from fastapi import FastAPI

    w = FastAPI()

    @w.get("/ping")
    def ping():
        return {"pong": True}

    return w


@app.function()
@modal.wsgi_app(label="wsgi")
def wsgi_fn():
    def application(environ, start_response):
        start_response("200 OK", [("Content-Type", "text/plain")])
        return [b"wsgi-ok"]

    return application


@app.cls()
class Greeter:
    @modal.enter()
    def setup(self):
        self.msg = "hi"

    @modal.fastapi_endpoint(method="GET", label="greet")
    def greet(self, name: str = "w"):
        return {"m": f"{self.msg}-{name}"}


def _client():
    root = build_ingress_app(app)
    return httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                             base_url="http://t")


def test_fastapi_get_with_query():
    async def go():
        async with _client() as c:
            r = await c.get("/double", params={"x": 21})
            assert r.json() == {"doubled": 42}

    asyncio.run(go())


def test_fastapi_post_with_body():
    async def go():
        async with _client() as c:
            r = await c.post("/concat", json={"a": "mi", "b": "355x"})
            assert r.json() == {"joined": "mi355x"}

    asyncio.run(go())


def test_asgi_mount():
    async def go():
        async with _client() as c:
            r = await c.get("/sub/ping")
            assert r.json() == {"pong": True}

    asyncio.run(go())


def test_wsgi_mount():
    async def go():
        async with _client() as c:
            r = await c.get("/wsgi/anything")
            assert r.text == "wsgi-ok"

    asyncio.run(go())


def test_cls_endpoint_runs_enter():
    async def go():
        async with _client() as c:
            r = await c.get("/greet", params={"name": "q"})
            assert r.json() == {"m": "hi-q"}

    asyncio.run(go())


def test_web_url_label():
    assert "double" in double.get_web_url()


app_sticky = modal.App("test-web-sticky")


@app_sticky.function(sticky=True, max_containers=4)
@modal.fastapi_endpoint(method="GET", label="whoami")
def whoami():
    import os

    return {"pid": os.getpid()}


def test_sticky_session_pins_worker():
    """sticky=True: requests sharing a Modal-Session header are served by one
    container (Modal Server sticky routing)."""
    root = build_ingress_app(app_sticky)

    async def go():
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            pids = set()
            for _ in range(4):
                r = await c.get("/whoami", headers={"Modal-Session": "s-1"})
                assert r.status_code == 200
                pids.add(r.json()["pid"])
            assert len(pids) == 1

    asyncio.run(go())


def test_proxy_auth_enforced(monkeypatch):
    """requires_proxy_auth: 401 without Modal-Key/Modal-Secret headers, 200
    with the keystore credentials; locked when no token is configured."""
    app_auth = modal.App("test-web-auth")

    @app_auth.function()
    @modal.fastapi_endpoint(method="GET", label="secret-square",
                            requires_proxy_auth=True)
    def ssq(x: int = 2):
        return {"sq": x * x}

    root = build_ingress_app(app_auth)

    async def go():
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            # no token configured: endpoint stays locked
            monkeypatch.delenv("MODAL_AMD_PROXY_TOKEN_ID", raising=False)
            r = await c.get("/secret-square", params={"x": 3})
            assert r.status_code == 401
            monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_ID", "wk-1")
            monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_SECRET", "ws-2")
            r = await c.get("/secret-square", params={"x": 3})
            assert r.status_code == 401  # still no headers
            r = await c.get("/secret-square", params={"x": 3},
                            headers={"Modal-Key": "wk-1", "Modal-Secret": "ws-2"})
            assert r.status_code == 200 and r.json() == {"sq": 9}
            r = await c.get("/secret-square",
                            headers={"Modal-Key": "wk-1", "Modal-Secret": "bad"})
            assert r.status_code == 401

    asyncio.run(go())


def test_proxy_auth_enforced_all_kinds(monkeypatch):
    """requires_proxy_auth applies to asgi/wsgi mounts, web_server helper
    routes, and class-based routes — not only function-level fastapi routes
    (ADVICE r1 high finding)."""
    app_auth2 = modal.App("test-web-auth-kinds")

    @app_auth2.function()
    @modal.asgi_app(label="locked-sub", requires_proxy_auth=True)
    def locked_sub():
        from fastapi import FastAPI

        w = FastAPI()

        @w.get("/ping")
        def ping():
            return {"pong": True}

        return w

    @app_auth2.function()
    @modal.wsgi_app(label="locked-wsgi", requires_proxy_auth=True)
    def locked_wsgi():
        def application(environ, start_response):
            start_response("200 OK", [("Content-Type", "text/plain")])
            return [b"wsgi-ok"]

        return application

    @app_auth2.cls()
    class LockedGreeter:
        @modal.fastapi_endpoint(method="GET", label="locked-greet",
                                requires_proxy_auth=True)
        def greet(self, name: str = "w"):
            return {"m": name}

    root = build_ingress_app(app_auth2)
    ok_headers = {"Modal-Key": "wk-1", "Modal-Secret": "ws-2"}

    async def go():
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_ID", "wk-1")
            monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_SECRET", "ws-2")
            for path in ("/locked-sub/ping", "/locked-wsgi/x", "/locked-greet"):
                r = await c.get(path)
                assert r.status_code == 401, path
            r = await c.get("/locked-sub/ping", headers=ok_headers)
            assert r.status_code == 200 and r.json() == {"pong": True}
            r = await c.get("/locked-wsgi/x", headers=ok_headers)
            assert r.status_code == 200 and r.text == "wsgi-ok"
            r = await c.get("/locked-greet", params={"name": "q"},
                            headers=ok_headers)
            assert r.status_code == 200 and r.json() == {"m": "q"}

    asyncio.run(go())


def test_websocket_through_asgi_mount():
    """WebSocket serving: a WS route on a mounted @modal.asgi_app works
    through the ingress (streaming-audio family dependency,
    streaming_kyutai_stt.py:334-390)."""
    app_ws = modal.App("test-web-ws")

    @app_ws.function()
    @modal.asgi_app(label="wsapp")
    def wsapp():
        from fastapi import FastAPI, WebSocket

        w = FastAPI()

        @w.websocket("/echo")
        async def echo(ws: WebSocket):
            await ws.accept()
            while True:
                msg = await ws.receive_text()
                if msg == "bye":
                    break
                await ws.send_text(msg.upper())
            await ws.close()

        return w

    from starlette.testclient import TestClient

    root = build_ingress_app(app_ws)
    with TestClient(root) as c:
        with c.websocket_connect("/wsapp/echo") as ws:
            ws.send_text("hello")
            assert ws.receive_text() == "HELLO"
            ws.send_text("bye")


def test_websocket_proxy_auth(monkeypatch):
    """A locked asgi mount refuses the WS handshake without credentials."""
    app_ws2 = modal.App("test-web-ws-auth")

    @app_ws2.function()
    @modal.asgi_app(label="lockedws", requires_proxy_auth=True)
    def lockedws():
        from fastapi import FastAPI, WebSocket

        w = FastAPI()

        @w.websocket("/s")
        async def s(ws: WebSocket):
            await ws.accept()
            await ws.send_text("in")
            await ws.close()

        return w

    from starlette.testclient import TestClient

    monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_ID", "k")
    monkeypatch.setenv("MODAL_AMD_PROXY_TOKEN_SECRET", "s")
    root = build_ingress_app(app_ws2)
    with TestClient(root) as c:
        import pytest as _pytest

        with _pytest.raises(Exception):
            with c.websocket_connect("/lockedws/s"):
                pass
        with c.websocket_connect(
                "/lockedws/s", headers={"Modal-Key": "k", "Modal-Secret": "s"}) as ws:
            assert ws.receive_text() == "in"
