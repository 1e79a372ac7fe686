"""CLI: ``run`` / ``serve`` / ``deploy`` / ``shell`` over example files.

Mirrors the reference's invocation surface (hello_world.py:73 ``modal run``,
text_to_image.py:157-163 auto-derived entrypoint flags incl. ``--flag/--no-flag``
booleans, flux.py:170-176).  Usage::

    python -m modal_examples_amd run examples/01_getting_started/hello_world.py
    python -m modal_examples_amd run file.py::entry --arg 3
    python -m modal_examples_amd serve examples/07_web/basic_web.py
    python -m modal_examples_amd deploy examples/05_scheduling/schedule_simple.py
    python -m modal_examples_amd app list            # deployed apps + heartbeats
    python -m modal_examples_amd shell file.py::Svc  # attach into a live worker
"""
from __future__ import annotations

import argparse
import importlib.util
import inspect
import os
import signal
import sys
import time
from pathlib import Path


def load_module(path: str):
    p = Path(path)
    spec = importlib.util.spec_from_file_location(p.stem.replace("-", "_"), p)
    mod = importlib.util.module_from_spec(spec)
    sys.modules[spec.name] = mod
    spec.loader.exec_module(mod)
    # file-loaded modules are not importable by worker processes — ship their
    # functions (and module globals they close over) by value instead
    import cloudpickle

    cloudpickle.register_pickle_by_value(mod)
    return mod


def find_app(mod):
    from .app import App

    for v in vars(mod).values():
        if isinstance(v, App):
            return v
    raise SystemExit(f"no modal App found in {mod.__name__}")


def _add_args_from_signature(parser: argparse.ArgumentParser, fn):
    for name, p in inspect.signature(fn).parameters.items():
        flag = "--" + name.replace("_", "-")
        ann = p.annotation if p.annotation is not inspect.Parameter.empty else str
        default = None if p.default is inspect.Parameter.empty else p.default
        required = p.default is inspect.Parameter.empty
        if ann is bool or isinstance(default, bool):
            group = parser.add_mutually_exclusive_group(required=False)
            group.add_argument(flag, dest=name, action="store_true")
            group.add_argument("--no-" + name.replace("_", "-"), dest=name,
                               action="store_false")
            parser.set_defaults(**{name: bool(default)})
        else:
            typ = ann if ann in (int, float, str) else str
            parser.add_argument(flag, dest=name, type=typ, default=default,
                                required=required)


def cmd_run(target: str, extra_args):
    path, _, entry = target.partition("::")
    mod = load_module(path)
    app = find_app(mod)
    if not app.entrypoints:
        raise SystemExit(f"app {app.name!r} has no @app.local_entrypoint")
    name = entry or next(iter(app.entrypoints))
    fn = app.entrypoints[name]
    parser = argparse.ArgumentParser(prog=f"run {path}::{name}")
    _add_args_from_signature(parser, fn)
    ns = parser.parse_args(extra_args)
    from .app import _Runtime

    try:
        fn(**vars(ns))
    finally:
        _Runtime.reset()


def cmd_serve(target: str, extra_args):
    parser = argparse.ArgumentParser(prog="serve")
    parser.add_argument("--port", type=int, default=8787)
    parser.add_argument("--timeout", type=float, default=0,
                        help="exit after N seconds (MODAL_SERVE_TIMEOUT analog)")
    ns = parser.parse_args(extra_args)
    import os

    timeout = ns.timeout or float(os.environ.get("MODAL_SERVE_TIMEOUT", 0))
    mod = load_module(target)
    app = find_app(mod)
    from .web.ingress import serve, stop_serving

    url = serve(app, port=ns.port, block=False)
    print(f"serving {app.name} at {url}")
    for f in list(app.web_endpoints.values()):
        print(f"  → {url}/{getattr(f.raw, '_modal_flags', {}).get('label') or f.name}")

    # hot reload (the reference `modal serve` watches the file and redeploys
    # on change): poll the target's mtime; on edit, tear down the ingress +
    # pools, re-import the module, serve the fresh app on the same port.
    import os as _os
    import sys as _sys

    path = target.split("::")[0]
    mtime = _os.path.getmtime(path) if _os.path.exists(path) else 0
    deadline = time.monotonic() + timeout if timeout else None
    try:
        while deadline is None or time.monotonic() < deadline:
            time.sleep(0.5)
            try:
                m = _os.path.getmtime(path)
            except OSError:
                continue
            if m != mtime:
                mtime = m
                print(f"⟳ {path} changed — reloading")
                stop_serving()
                # wait for the old uvicorn to release the port (rebind race)
                import socket as _socket

                for _ in range(50):
                    s_ = _socket.socket()
                    try:
                        s_.bind(("127.0.0.1", ns.port))
                        s_.close()
                        break
                    except OSError:
                        s_.close()
                        time.sleep(0.1)
                from .app import _Runtime

                _Runtime.reset()
                for name in [n for n, mod_ in list(_sys.modules.items())
                             if getattr(mod_, "__file__", None)
                             and _os.path.abspath(str(getattr(mod_, "__file__")))
                             == _os.path.abspath(path)]:
                    del _sys.modules[name]
                try:
                    mod = load_module(target)
                    app = find_app(mod)
                    url = serve(app, port=ns.port, block=False)
                    print(f"re-serving {app.name} at {url}")
                except Exception as e:  # noqa: BLE001 — keep watching
                    print(f"reload failed (fix the file and save again): {e}")
    except KeyboardInterrupt:
        pass
    finally:
        stop_serving()
        from .app import _Runtime

        _Runtime.reset()


def cmd_deploy(target: str, extra_args):
    mod = load_module(target)
    app = find_app(mod)
    app.deploy()
    from . import config

    rec = config.state_dir() / "deployments.txt"
    with open(rec, "a") as f:
        f.write(f"{time.strftime('%F %T')} {app.name} {target}\n")
    scheduled = [n for n, fn in app.functions.items() if fn.opts.schedule]
    print(f"deployed {app.name} ({len(app.functions)} functions, "
          f"{len(app.classes)} classes)")
    if scheduled:
        print(f"schedules active for: {', '.join(scheduled)} — keeping process alive")
        try:
            while True:
                time.sleep(3600)
        except KeyboardInterrupt:
            pass


def cmd_shell(target: str, extra_args):
    """``shell file.py`` → local REPL over the module; ``shell
    file.py::FuncOrCls`` → attach to a RUNNING container of that service
    (warming one if needed) and exec lines inside the worker process, where
    the live models/caches are loaded."""
    import code

    if target and "::" in target:
        path, name = target.split("::", 1)
        mod = load_module(path)
        app = find_app(mod)
        fn = app.functions.get(name)
        if fn is None and name in app.classes:
            c = app.classes[name]()
            fn = next(m for m in (getattr(c, n, None) for n in dir(c))
                      if hasattr(m, "_submit"))
        if fn is None:
            raise SystemExit(f"no function or class {name!r} in {path}")
        pool = fn._get_pool() if hasattr(fn, "_get_pool") else fn.obj._get_pool()
        print(f"attached to a worker of {name} — `obj` is the live instance, "
              "`methods` the service methods; blank line or Ctrl-D exits.")
        while True:
            try:
                line = input(f"({name}) >>> ")
            except EOFError:
                break
            if not line.strip():
                break
            try:
                out = pool.submit("__debug_exec__", (line,), {},
                                  sticky_key="__shell__").wait()
                if out:
                    print(out, end="" if out.endswith("\n") else "\n")
            except BaseException as e:  # noqa: BLE001
                print(f"{type(e).__name__}: {e}")
        return
    mod = load_module(target) if target else None
    banner = "modal_examples_amd shell"
    ns = dict(vars(mod)) if mod else {}
    code.interact(banner=banner, local=ns)


def cmd_volume(args):
    """volume ls|get|put|rm <name> [paths] — the `modal volume` surface."""
    from .resources.volume import Volume

    sub, name, *rest = args
    vol = Volume.from_name(name, create_if_missing=True)
    if sub == "ls":
        for f in sorted(vol.listdir(rest[0] if rest else "/")):
            print(f)
    elif sub == "get":
        remote, local = rest[0], (rest[1] if len(rest) > 1 else Path(rest[0]).name)
        Path(local).write_bytes(vol.read_file(remote))
        print(f"wrote {local}")
    elif sub == "put":
        local, remote = rest[0], (rest[1] if len(rest) > 1 else Path(rest[0]).name)
        dst = vol.path / remote.lstrip("/")
        dst.parent.mkdir(parents=True, exist_ok=True)
        dst.write_bytes(Path(local).read_bytes())
        vol.commit()
        print(f"put {remote}")
    elif sub == "rm":
        vol.remove_file(rest[0], recursive="-r" in rest)
        print(f"removed {rest[0]}")
    else:
        raise SystemExit(f"unknown volume subcommand {sub}")


def cmd_dict(args):
    from .resources.dict_queue import Dict

    sub, name, *rest = args
    d = Dict.from_name(name)
    if sub == "get":
        print(d.get(rest[0]))
    elif sub == "set":
        d[rest[0]] = rest[1]
        print("ok")
    elif sub == "items":
        for k, v in d.items():
            print(f"{k}\t{v}")
    elif sub == "clear":
        d.clear()


def cmd_queue(args):
    from .resources.dict_queue import Queue

    sub, name, *rest = args
    q = Queue.from_name(name)
    if sub == "len":
        print(q.len(rest[0] if rest else None))
    elif sub == "put":
        q.put(rest[0])
    elif sub == "get":
        print(q.get(block=False))
    elif sub == "clear":
        q.clear(all=True)


def cmd_secret(args):
    from .resources.secret import Secret

    sub, name, *rest = args
    if sub == "create":
        env = dict(kv.split("=", 1) for kv in rest)
        Secret.create(name, env)
        print(f"created secret {name} with {len(env)} keys")


def cmd_app(args):
    """app list|stop <name> — deployed-app management (the `modal app`
    surface): list shows heartbeat state, stop SIGTERMs the deploy process
    and clears the record."""
    from .runtime.store import DictStore

    d = DictStore("__deployments__")
    sub = args[0] if args else "list"
    if sub == "list":
        now = time.time()
        items = sorted(d.items())
        if not items:
            print("no deployments")
            return
        for name, beat in items:
            ts = beat.get("t") if isinstance(beat, dict) else beat
            pid = beat.get("pid") if isinstance(beat, dict) else None
            age = now - (ts or 0)
            state = "live" if age < 30 else f"stale {int(age)}s"
            print(f"{name:36s} {state:14s} pid={pid or '?'}")
    elif sub == "stop":
        name = args[1]
        beat = d.get(name)
        pid = beat.get("pid") if isinstance(beat, dict) else None
        if pid:
            try:
                os.kill(int(pid), signal.SIGTERM)
                print(f"sent SIGTERM to {name} (pid {pid})")
            except (ProcessLookupError, PermissionError):
                print(f"{name}: deploy process already gone")
        d.delete(name)
        print(f"cleared deployment record for {name}")
    else:
        raise SystemExit("usage: app list | app stop <name>")


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    if not argv or argv[0] in ("-h", "--help"):
        print(__doc__)
        return 0
    cmd, *rest = argv
    if cmd == "run":
        cmd_run(rest[0], rest[1:])
    elif cmd == "serve":
        cmd_serve(rest[0], rest[1:])
    elif cmd == "deploy":
        cmd_deploy(rest[0], rest[1:])
    elif cmd == "shell":
        cmd_shell(rest[0] if rest else None, rest[1:])
    elif cmd == "volume":
        cmd_volume(rest)
    elif cmd == "dict":
        cmd_dict(rest)
    elif cmd == "queue":
        cmd_queue(rest)
    elif cmd == "secret":
        cmd_secret(rest)
    elif cmd == "app":
        cmd_app(rest)
    elif cmd in ("--version", "version"):
        from . import __version__

        print(f"modal_examples_amd {__version__} (MI355X / ROCm)")
    else:
        raise SystemExit(
            f"unknown command {cmd!r}; use run/serve/deploy/shell/app/"
            "volume/dict/queue/secret")
    return 0
