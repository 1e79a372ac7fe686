"""Local S3-compatible object store: REST endpoint + client + prefix sync.

Gives CloudBucketMount real bucket semantics (reference:
10_integrations/s3_bucket_mount.py:63-80) on a no-network node: an ASGI app
speaking the S3 REST subset (ListObjectsV2, GET/PUT/HEAD/DELETE object)
backed by the state dir, a dependency-free HTTP client, and prefix-sync
helpers the worker uses to materialize a mount (download-on-mount,
write-back-on-exit) instead of symlinking a shared directory.
"""
from __future__ import annotations

import threading
import urllib.parse
import urllib.request
from pathlib import Path
from xml.sax.saxutils import escape

from .. import config

_server = {"port": None, "thread": None}
_lock = threading.Lock()


def bucket_root() -> Path:
    return config.state_dir() / "buckets"


def make_s3_app():
    """ASGI app implementing the S3 REST subset over the bucket root."""

    async def app(scope, receive, send):
        if scope["type"] != "http":
            return
        method = scope["method"]
        raw = scope["path"].lstrip("/")
        qs = urllib.parse.parse_qs(scope.get("query_string", b"").decode())
        parts = raw.split("/", 1)
        bucket = urllib.parse.unquote(parts[0])
        key = urllib.parse.unquote(parts[1]) if len(parts) > 1 else ""
        root = bucket_root() / bucket

        def path_ok() -> bool:
            # reject traversal: the resolved object path must stay inside
            # the bucket root (and bucket names must be plain)
            if "/" in bucket or ".." in bucket or bucket.startswith("."):
                return False
            if not key:
                return True
            full = (root / key).resolve()
            return str(full).startswith(str(root.resolve()) + "/") or \
                full == root.resolve()

        async def respond(status, body=b"", ctype="application/xml"):
            await send({"type": "http.response.start", "status": status,
                        "headers": [(b"content-type", ctype.encode()),
                                    (b"content-length", str(len(body)).encode())]})
            await send({"type": "http.response.body", "body": body})

        if not bucket or not path_ok():
            await respond(400, b"<Error><Code>InvalidKey</Code></Error>")
            return

        if method == "GET" and (not key or "list-type" in qs):
            # ListObjectsV2
            prefix = qs.get("prefix", [""])[0]
            items = []
            if root.exists():
                for f in sorted(root.glob("**/*")):
                    if f.is_file():
                        k = str(f.relative_to(root))
                        if k.startswith(prefix):
                            items.append((k, f.stat().st_size))
            xml = ["<?xml version=\"1.0\"?><ListBucketResult>",
                   f"<Name>{escape(bucket)}</Name>",
                   f"<Prefix>{escape(prefix)}</Prefix>",
                   f"<KeyCount>{len(items)}</KeyCount>"]
            for k, size in items:
                xml.append(f"<Contents><Key>{escape(k)}</Key>"
                           f"<Size>{size}</Size></Contents>")
            xml.append("</ListBucketResult>")
            await respond(200, "".join(xml).encode())
            return

        path = root / key
        if method in ("GET", "HEAD"):
            if not path.is_file():
                await respond(404, b"<Error><Code>NoSuchKey</Code></Error>")
                return
            body = b"" if method == "HEAD" else path.read_bytes()
            await respond(200, body, "application/octet-stream")
        elif method == "PUT":
            body = b""
            while True:
                msg = await receive()
                body += msg.get("body", b"")
                if not msg.get("more_body"):
                    break
            path.parent.mkdir(parents=True, exist_ok=True)
            path.write_bytes(body)
            await respond(200, b"")
        elif method == "DELETE":
            if path.is_file():
                path.unlink()
            await respond(204, b"")
        else:
            await respond(405, b"<Error><Code>MethodNotAllowed</Code></Error>")

    return app


def start_s3_server() -> str:
    """Start (once per process) the local S3 endpoint; returns its URL."""
    with _lock:
        if _server["port"] is None:
            import socket

            import uvicorn

            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
            s.close()
            cfg = uvicorn.Config(make_s3_app(), host="127.0.0.1", port=port,
                                 log_level="warning")
            srv = uvicorn.Server(cfg)
            t = threading.Thread(target=srv.run, daemon=True)
            t.start()
            import time

            deadline = time.monotonic() + 10
            while not srv.started and time.monotonic() < deadline:
                time.sleep(0.02)
            _server.update(port=port, thread=t)
        return f"http://127.0.0.1:{_server['port']}"


class S3Client:
    """Dependency-free client for the S3 REST subset."""

    def __init__(self, endpoint: str):
        self.endpoint = endpoint.rstrip("/")

    def _url(self, bucket: str, key: str = "", query: str = "") -> str:
        u = f"{self.endpoint}/{urllib.parse.quote(bucket)}"
        if key:
            u += "/" + urllib.parse.quote(key)
        if query:
            u += "?" + query
        return u

    def list(self, bucket: str, prefix: str = "") -> list:
        q = "list-type=2&prefix=" + urllib.parse.quote(prefix)
        with urllib.request.urlopen(self._url(bucket, query=q), timeout=10) as r:
            text = r.read().decode()
        import re

        return re.findall(r"<Key>(.*?)</Key>", text)

    def get(self, bucket: str, key: str) -> bytes:
        with urllib.request.urlopen(self._url(bucket, key), timeout=30) as r:
            return r.read()

    def put(self, bucket: str, key: str, data: bytes) -> None:
        req = urllib.request.Request(self._url(bucket, key), data=data,
                                     method="PUT")
        urllib.request.urlopen(req, timeout=30).read()

    def delete(self, bucket: str, key: str) -> None:
        req = urllib.request.Request(self._url(bucket, key), method="DELETE")
        urllib.request.urlopen(req, timeout=10).read()

    # ---- prefix sync (the mount materialization) ----

    def sync_down(self, bucket: str, prefix: str, dest: Path) -> int:
        dest = Path(dest)
        n = 0
        for key in self.list(bucket, prefix):
            rel = key[len(prefix):].lstrip("/") if prefix else key
            p = dest / rel
            p.parent.mkdir(parents=True, exist_ok=True)
            p.write_bytes(self.get(bucket, key))
            n += 1
        return n

    def sync_up(self, bucket: str, prefix: str, src: Path) -> int:
        src = Path(src)
        n = 0
        if not src.exists():
            return 0
        for f in src.glob("**/*"):
            if f.is_file():
                key = (prefix.rstrip("/") + "/" if prefix else "") + str(
                    f.relative_to(src))
                self.put(bucket, key, f.read_bytes())
                n += 1
        return n
