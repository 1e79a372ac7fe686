"""Web layer: endpoint decorators + the local ingress.

Reference surface: ``@modal.fastapi_endpoint`` (25 uses),
``@modal.asgi_app`` (39), ``@modal.wsgi_app`` (5), ``@modal.web_server`` (19),
``@app.server`` (31), sticky routing (07_web/server_sticky.py), proxy auth
(07_web/basic_web.py:178-180), tunnels (11_notebooks/jupyter_inside_modal.py:60).

Local design: one uvicorn ingress in the client process routes HTTP to the
function's worker pool (``.remote`` under the hood), mirroring Modal's
ingress→container hop.  ``web_server``/``@app.server`` functions bind their own
port in the worker; the ingress health-checks the port and hands out the URL
(503-until-warm loop semantics, 07_web/server.py:92-110).
"""
from __future__ import annotations

import asyncio
import socket
import threading
import time
from typing import Optional

from ..app import _set_flag

_DEFAULT_PORT = 8787
_ingress_state = {"port": None, "server": None, "apps": []}


# ---------------------------------------------------------------- decorators


def fastapi_endpoint(method: str = "GET", label: Optional[str] = None,
                     docs: bool = False, custom_domains=None,
                     requires_proxy_auth: bool = False):
    def deco(fn):
        return _set_flag(fn, web=True, web_kind="fastapi", http_method=method,
                         label=label or getattr(fn, "__name__", "web"),
                         requires_proxy_auth=requires_proxy_auth)
    return deco


# legacy alias used by older examples
web_endpoint = fastapi_endpoint


def asgi_app(label: Optional[str] = None, requires_proxy_auth: bool = False):
    def deco(fn):
        return _set_flag(fn, web=True, web_kind="asgi", label=label,
                         requires_proxy_auth=requires_proxy_auth)
    return deco


def wsgi_app(label: Optional[str] = None, requires_proxy_auth: bool = False):
    def deco(fn):
        return _set_flag(fn, web=True, web_kind="wsgi", label=label,
                         requires_proxy_auth=requires_proxy_auth)
    return deco


def web_server(port: int, startup_timeout: float = 60.0, label: Optional[str] = None,
               custom_domains=None, requires_proxy_auth: bool = False):
    def deco(fn):
        return _set_flag(fn, web=True, web_kind="server", port=port,
                         startup_timeout=startup_timeout, label=label,
                         requires_proxy_auth=requires_proxy_auth)
    return deco


class forward:
    """``modal.forward(port)`` tunnel context manager — locally a loopback URL
    (jupyter_inside_modal.py:57-83)."""

    def __init__(self, port: int, unencrypted: bool = False):
        self.port = port

    def __enter__(self):
        from ..runtime.sandbox import Tunnel

        return Tunnel("127.0.0.1", self.port)

    def __exit__(self, *a):
        return False


# ---------------------------------------------------------------- ingress


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def web_url_for(fn) -> str:
    flags = getattr(fn.raw, "_modal_flags", {}) if hasattr(fn, "raw") else {}
    port = _ingress_state["port"] or _DEFAULT_PORT
    label = flags.get("label") or getattr(fn, "name", "web")
    return f"http://127.0.0.1:{port}/{label}"


def _proxy_auth_ok(request) -> bool:
    """Proxy-auth check (basic_web.py:178-180 pattern): token id/secret via
    Modal-Key/Modal-Secret headers, verified against the local keystore env."""
    import os

    want_key = os.environ.get("MODAL_AMD_PROXY_TOKEN_ID", "")
    want_secret = os.environ.get("MODAL_AMD_PROXY_TOKEN_SECRET", "")
    if not want_key:
        return False  # locked endpoints stay locked until a token is set
    return (request.headers.get("Modal-Key") == want_key and
            request.headers.get("Modal-Secret", "") == want_secret)


class _ProxyAuthASGI:
    """ASGI middleware enforcing proxy auth on mounted asgi/wsgi sub-apps —
    the locked-endpoint contract of basic_web.py:178-180 applies to every
    web kind, not just function-level fastapi routes."""

    def __init__(self, inner):
        self.inner = inner

    async def __call__(self, scope, receive, send):
        if scope["type"] in ("http", "websocket"):
            headers = {k.decode("latin1").lower(): v.decode("latin1")
                       for k, v in scope.get("headers", [])}
            if not _proxy_auth_headers_ok(headers):
                if scope["type"] == "websocket":
                    await send({"type": "websocket.close", "code": 4401})
                    return
                await send({"type": "http.response.start", "status": 401,
                            "headers": [(b"content-type", b"application/json")]})
                await send({"type": "http.response.body",
                            "body": b'{"detail": "proxy auth required"}'})
                return
        await self.inner(scope, receive, send)


def _proxy_auth_headers_ok(headers: dict) -> bool:
    import os

    want_key = os.environ.get("MODAL_AMD_PROXY_TOKEN_ID", "")
    want_secret = os.environ.get("MODAL_AMD_PROXY_TOKEN_SECRET", "")
    if not want_key:
        return False
    return (headers.get("modal-key") == want_key and
            headers.get("modal-secret", "") == want_secret)


def build_ingress_app(app):
    """Build one FastAPI app exposing every web endpoint of ``app``."""
    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse

    root = FastAPI(title=f"{app.name} (local MI355X ingress)")

    def _add_function_route(f, flags):
        kind = flags.get("web_kind")
        label = flags.get("label") or f.name
        if kind == "fastapi":
            method = flags.get("http_method", "GET")
            # run the user function in its worker pool, forwarding query/body kwargs
            import inspect as _inspect

            sig = _inspect.signature(f.raw)
            target = f  # closed over — FastAPI deep-copies handler DEFAULTS,
            # and Function objects hold locks

            needs_auth = bool(flags.get("requires_proxy_auth"))
            is_sticky = bool(getattr(f.opts, "sticky", False))

            async def handler(request: Request):
                if needs_auth and not _proxy_auth_ok(request):
                    return JSONResponse({"detail": "proxy auth required"},
                                        status_code=401)
                kwargs = dict(request.query_params)
                if request.method in ("POST", "PUT"):
                    try:
                        body = await request.json()
                        if isinstance(body, dict):
                            kwargs.update(body)
                    except Exception:
                        pass
                # coerce types using the signature annotations
                coerced = {}
                for name, p in sig.parameters.items():
                    if name in kwargs:
                        v = kwargs[name]
                        if p.annotation in (int, float, bool) and isinstance(v, str):
                            v = p.annotation(v) if p.annotation is not bool else v.lower() in ("1", "true", "yes")
                        coerced[name] = v
                if is_sticky:
                    # Modal Server sticky routing: all requests of one client
                    # session land on one container.  Session = Modal-Session
                    # header, else the client address.
                    key = request.headers.get("Modal-Session") or (
                        request.client.host if request.client else "anon")
                    result = await asyncio.to_thread(
                        lambda: target._submit((), coerced, sticky_key=key).wait())
                else:
                    result = await asyncio.to_thread(target.remote, **coerced)
                if hasattr(result, "__class__") and result.__class__.__name__ == "Response":
                    return result
                if isinstance(result, (bytes, bytearray)):
                    from fastapi.responses import Response

                    return Response(content=bytes(result))
                return JSONResponse(result) if not hasattr(result, "status_code") else result

            # module uses `from __future__ import annotations`: give FastAPI a
            # REAL class, not the string "Request"
            handler.__annotations__ = {"request": Request}
            root.add_api_route(f"/{label}", handler, methods=[method])
        elif kind == "asgi":
            sub = f.raw()  # factory runs in-process
            if flags.get("requires_proxy_auth"):
                sub = _ProxyAuthASGI(sub)
            root.mount(f"/{label}" if label else "", sub)
        elif kind == "wsgi":
            from starlette.middleware.wsgi import WSGIMiddleware

            sub = WSGIMiddleware(f.raw())
            if flags.get("requires_proxy_auth"):
                sub = _ProxyAuthASGI(sub)
            root.mount(f"/{label}" if label else "", sub)
        elif kind == "server":
            port = flags.get("port")
            needs_auth = bool(flags.get("requires_proxy_auth"))
            f.spawn()  # starts the server inside a worker

            @root.get(f"/{label}/_url")
            async def url_handler(request: Request, _port=port,
                                  _auth=needs_auth):
                if _auth and not _proxy_auth_ok(request):
                    return JSONResponse({"detail": "proxy auth required"},
                                        status_code=401)
                return {"url": f"http://127.0.0.1:{_port}"}

    for f in app.web_endpoints.values():
        flags = getattr(f.raw, "_modal_flags", {})
        _add_function_route(f, flags)
    for c in app.classes.values():
        for mname in dir(c.user_cls):
            m = getattr(c.user_cls, mname, None)
            flags = getattr(m, "_modal_flags", None)
            if flags and flags.get("web"):
                _add_cls_route(root, c, mname, flags)
    return root


def _add_cls_route(root, c, mname, flags):
    import inspect as _inspect

    from fastapi import Request
    from fastapi.responses import JSONResponse

    kind = flags.get("web_kind")
    label = flags.get("label") or mname
    obj = c()

    needs_auth = bool(flags.get("requires_proxy_auth"))

    if kind == "fastapi":
        sig = _inspect.signature(getattr(c.user_cls, mname))
        method = flags.get("http_method", "GET")

        async def handler(request: Request):
            if needs_auth and not _proxy_auth_ok(request):
                return JSONResponse({"detail": "proxy auth required"},
                                    status_code=401)
            kwargs = dict(request.query_params)
            if request.method in ("POST", "PUT"):
                try:
                    body = await request.json()
                    if isinstance(body, dict):
                        kwargs.update(body)
                except Exception:
                    pass
            coerced = {}
            for name, p in sig.parameters.items():
                if name == "self":
                    continue
                if name in kwargs:
                    v = kwargs[name]
                    if p.annotation in (int, float, bool) and isinstance(v, str):
                        v = p.annotation(v) if p.annotation is not bool else v.lower() in ("1", "true", "yes")
                    coerced[name] = v
            bm = getattr(obj, mname)
            result = await asyncio.to_thread(bm.remote, **coerced)
            if isinstance(result, (bytes, bytearray)):
                from fastapi.responses import Response

                return Response(content=bytes(result))
            return JSONResponse(result)

        handler.__annotations__ = {"request": Request}
        root.add_api_route(f"/{label}", handler, methods=[method])
    elif kind == "asgi":
        inst = obj._local_instance()
        sub = getattr(inst, mname)()
        if needs_auth:
            sub = _ProxyAuthASGI(sub)
        root.mount(f"/{label}" if label else "", sub)


def serve(app, port: Optional[int] = None, block: bool = False) -> str:
    """Start the local ingress for an app. Returns base URL."""
    import uvicorn

    port = port or _DEFAULT_PORT
    root = build_ingress_app(app)
    cfg = uvicorn.Config(root, host="127.0.0.1", port=port, log_level="warning")
    server = uvicorn.Server(cfg)
    _ingress_state["port"] = port
    _ingress_state["server"] = server
    if block:
        server.run()
    else:
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        deadline = time.monotonic() + 15
        while not server.started and time.monotonic() < deadline:
            time.sleep(0.05)
    return f"http://127.0.0.1:{port}"


def stop_serving():
    s = _ingress_state.get("server")
    if s is not None:
        s.should_exit = True
        _ingress_state["server"] = None
