"""Sandbox: isolated subprocess execution environment.

Reference API surface: 13_sandboxes/safe_code_execution.py:28-46 (create + exec
with stdout/stderr streams), sandbox_pool.py:137-292 (warm pool, readiness
probes, TTL).  Locally a sandbox is a process group rooted in a scratch
directory; ``exec`` spawns processes inside it.  (No gVisor-grade isolation —
documented non-goal, SURVEY.md §7 "out of scope".)
"""
from __future__ import annotations

import os
import shutil
import signal
import subprocess
import tempfile
import threading
import time
import uuid
from typing import List, Optional

from ..exception import SandboxTimeoutError


class _Stream:
    def __init__(self, f):
        self._f = f

    def read(self) -> str:
        data = self._f.read()
        return data.decode() if isinstance(data, bytes) else (data or "")

    def __iter__(self):
        for line in self._f:
            yield line.decode() if isinstance(line, bytes) else line


class ContainerProcess:
    def __init__(self, popen: subprocess.Popen):
        self._p = popen
        self.stdout = _Stream(popen.stdout)
        self.stderr = _Stream(popen.stderr)
        self.stdin = popen.stdin

    def wait(self) -> int:
        return self._p.wait()

    def poll(self) -> Optional[int]:
        return self._p.poll()

    @property
    def returncode(self):
        return self._p.returncode

    def kill(self):
        self._p.kill()


class Probe:
    """Readiness probe (sandbox_pool.py:149)."""

    def __init__(self, kind: str, args: List[str]):
        self.kind = kind
        self.args = args

    @staticmethod
    def with_exec(args: List[str]) -> "Probe":
        return Probe("exec", list(args))


class Tunnel:
    def __init__(self, host: str, port: int):
        self.host = host
        self.port = port
        self.url = f"http://{host}:{port}"

    @property
    def tls_socket(self):
        return (self.host, self.port)


class Sandbox:
    _registry = {}

    def __init__(self, entrypoint_args, image=None, app=None, timeout: float = 600,
                 workdir: Optional[str] = None, volumes=None, encrypted_ports=None,
                 unencrypted_ports=None, cpu=None, memory=None, gpu=None,
                 readiness_probe: Optional[Probe] = None, env=None, **kw):
        self.object_id = "sb-" + uuid.uuid4().hex[:12]
        self.timeout = timeout
        self._root = tempfile.mkdtemp(prefix="mxa_sandbox_")
        self.workdir = workdir or self._root
        if workdir and not os.path.isdir(workdir):
            os.makedirs(workdir, exist_ok=True)
        self._env = dict(os.environ)
        self._env.update(env or {})
        # volumes: mount_path -> Volume: expose via env + symlink inside root
        for mnt, vol in (volumes or {}).items():
            target = getattr(vol, "path", None)
            if target is not None:
                link = os.path.join(self._root, mnt.lstrip("/"))
                os.makedirs(os.path.dirname(link) or self._root, exist_ok=True)
                if not os.path.exists(link):
                    os.symlink(target, link)
                if not os.path.exists(mnt):
                    try:
                        os.symlink(target, mnt)
                    except OSError:
                        pass
        self._procs: List[subprocess.Popen] = []
        self._main: Optional[ContainerProcess] = None
        self._created = time.monotonic()
        self._ports = list(encrypted_ports or []) + list(unencrypted_ports or [])
        self._terminated = False
        if entrypoint_args:
            self._main = self.exec(*entrypoint_args)
        self._reaper = threading.Timer(timeout, self._on_timeout)
        self._reaper.daemon = True
        self._reaper.start()
        Sandbox._registry[self.object_id] = self
        # cross-process registry: enough metadata for another client to
        # attach (exec/poll/terminate) — warm sandbox pools are shareable
        try:
            from . import store

            store.DictStore("__sandboxes__").put(self.object_id, {
                "workdir": self.workdir,
                "root": self._root,
                "main_pid": self._main._p.pid if self._main else None,
                "owner_pid": os.getpid(),
            })
        except Exception:
            pass
        self._probe = readiness_probe
        self._detached = False
        if readiness_probe is not None and readiness_probe.kind == "exec":
            self.wait_until_ready(min(60.0, timeout))

    def wait_until_ready(self, timeout: float = 60.0) -> bool:
        """Poll the readiness probe until it succeeds (sb.wait_until_ready,
        13_sandboxes usage).  True on ready; False on timeout/no probe."""
        if self._probe is None or self._probe.kind != "exec":
            return self.poll() is None
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            p = self.exec(*self._probe.args)
            if p.wait() == 0:
                return True
            time.sleep(0.5)
        return False

    def detach(self):
        """Detach from the app lifecycle: the sandbox (and its timeout reaper)
        keeps running after the creating context exits (sb.detach,
        13_sandboxes/opencode_server.py pattern); reattach via from_id."""
        self._detached = True
        return self

    # --- modal API ---

    @staticmethod
    def create(*entrypoint_args, **kw) -> "Sandbox":
        return Sandbox(list(entrypoint_args), **kw)

    @staticmethod
    def from_id(object_id: str) -> "Sandbox":
        """In-process handles come back whole; for a sandbox created by
        ANOTHER process a store-backed remote handle supports
        exec/poll/terminate (the warm-pool sharing pattern)."""
        s = Sandbox._registry.get(object_id)
        if s is not None:
            return s
        from . import store

        meta = store.DictStore("__sandboxes__").get(object_id)
        if meta is None:
            raise KeyError(f"sandbox {object_id} not found")
        return _RemoteSandbox(object_id, meta)

    @staticmethod
    def list(app=None):
        return [s for s in Sandbox._registry.values() if not s._terminated]

    def exec(self, *args, workdir: Optional[str] = None, timeout: Optional[float] = None,
             text: bool = True, **kw) -> ContainerProcess:
        p = subprocess.Popen(
            list(args),
            cwd=workdir or self.workdir,
            env=self._env,
            stdout=subprocess.PIPE,
            stderr=subprocess.PIPE,
            stdin=subprocess.PIPE,
            text=text,
            start_new_session=True,
        )
        self._procs.append(p)
        return ContainerProcess(p)

    def wait(self, raise_on_termination: bool = True) -> int:
        if self._main is None:
            return 0
        rc = self._main.wait()
        if self._terminated and raise_on_termination:
            raise SandboxTimeoutError(f"sandbox {self.object_id} terminated")
        return rc

    def poll(self) -> Optional[int]:
        if self._main is None:
            return 0 if self._terminated else None
        return self._main.poll()

    @property
    def returncode(self):
        return None if self._main is None else self._main.returncode

    @property
    def stdout(self):
        return self._main.stdout if self._main else _Stream(open(os.devnull))

    @property
    def stderr(self):
        return self._main.stderr if self._main else _Stream(open(os.devnull))

    def tunnels(self, timeout: float = 30):
        return {port: Tunnel("127.0.0.1", port) for port in self._ports}

    def open(self, path: str, mode: str = "r"):
        full = path if os.path.isabs(path) else os.path.join(self._root, path)
        os.makedirs(os.path.dirname(full) or "/", exist_ok=True)
        return open(full, mode)

    def mkdir(self, path: str, parents: bool = True):
        full = path if os.path.isabs(path) else os.path.join(self._root, path)
        os.makedirs(full, exist_ok=parents)

    def ls(self, path: str):
        full = path if os.path.isabs(path) else os.path.join(self._root, path)
        return os.listdir(full)

    def set_tags(self, tags: dict):
        self._tags = dict(tags)

    def terminate(self):
        self._terminated = True
        self._reaper.cancel()
        for p in self._procs:
            try:
                os.killpg(os.getpgid(p.pid), signal.SIGKILL)
            except Exception:
                try:
                    p.kill()
                except Exception:
                    pass
        shutil.rmtree(self._root, ignore_errors=True)
        Sandbox._registry.pop(self.object_id, None)

    def _on_timeout(self):
        self.terminate()


class _RemoteSandbox:
    """Cross-process view of a sandbox created elsewhere: exec runs in the
    same workdir, poll/terminate act on the recorded main pid."""

    def __init__(self, object_id: str, meta: dict):
        self.object_id = object_id
        self.workdir = meta.get("workdir")
        self._main_pid = meta.get("main_pid")
        self._env = dict(os.environ)
        self._terminated = False

    def exec(self, *args, workdir: Optional[str] = None, text: bool = True,
             **kw) -> ContainerProcess:
        p = subprocess.Popen(list(args), cwd=workdir or self.workdir,
                             env=self._env, stdout=subprocess.PIPE,
                             stderr=subprocess.PIPE, stdin=subprocess.PIPE,
                             text=text, start_new_session=True)
        return ContainerProcess(p)

    def poll(self) -> Optional[int]:
        if self._main_pid is None:
            return 0
        try:
            os.kill(self._main_pid, 0)
            return None  # still running
        except ProcessLookupError:
            return 0

    def terminate(self):
        self._terminated = True
        if self._main_pid:
            try:
                os.killpg(os.getpgid(self._main_pid), signal.SIGKILL)
            except Exception:
                try:
                    os.kill(self._main_pid, signal.SIGKILL)
                except Exception:
                    pass
