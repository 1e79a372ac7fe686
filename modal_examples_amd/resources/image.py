"""Image builder: a content-hashed recipe of environment layers.

The reference builds container images with a chained builder
(02_building_containers/install_cuda.py:40-42, stable_diffusion/text_to_image.py:47-65).
Locally there is one ROCm environment and no network, so the builder records the
layer chain (for provenance + content hash), applies what is locally applicable
(``.env`` vars are injected into worker processes; ``run_function`` executes once
at first use), and treats package-install layers as assertions that the package
is importable.
"""
from __future__ import annotations

import contextlib
import hashlib
import json
from typing import Callable, List, Optional


class Image:
    def __init__(self, base: str = "debian_slim", layers: Optional[List] = None):
        self.base = base
        self.layers: List = list(layers or [])
        self._env: dict = {}
        self._build_fns: List = []
        for kind, payload in self.layers:
            if kind == "env":
                self._env.update(payload)

    # ---- constructors ----

    @staticmethod
    def debian_slim(python_version: Optional[str] = None) -> "Image":
        return Image("debian_slim", [("python", python_version or "local")])

    @staticmethod
    def from_registry(tag: str, add_python: Optional[str] = None, **kw) -> "Image":
        return Image(f"registry:{tag}", [("python", add_python or "local")])

    @staticmethod
    def micromamba(python_version: Optional[str] = None) -> "Image":
        return Image("micromamba", [("python", python_version or "local")])

    # ---- chainable layers ----

    def _with(self, kind: str, payload) -> "Image":
        img = Image(self.base, self.layers + [(kind, payload)])
        img._build_fns = list(self._build_fns)
        if kind == "run_function":
            img._build_fns.append(payload)
        return img

    def pip_install(self, *pkgs, **kw) -> "Image":
        return self._with("pip", list(pkgs))

    def uv_pip_install(self, *pkgs, **kw) -> "Image":
        return self._with("pip", list(pkgs))

    def uv_sync(self, *a, **kw) -> "Image":
        return self._with("uv_sync", a)

    def apt_install(self, *pkgs, **kw) -> "Image":
        return self._with("apt", list(pkgs))

    def run_commands(self, *cmds, **kw) -> "Image":
        return self._with("cmd", list(cmds))

    def env(self, env: dict) -> "Image":
        return self._with("env", dict(env))

    def workdir(self, path: str) -> "Image":
        return self._with("workdir", path)

    def entrypoint(self, cmd: list) -> "Image":
        return self._with("entrypoint", list(cmd))

    def add_local_file(self, local_path, remote_path, copy: bool = False) -> "Image":
        return self._with("add_file", (str(local_path), str(remote_path)))

    def add_local_dir(self, local_path, remote_path, copy: bool = False, ignore=None) -> "Image":
        return self._with("add_dir", (str(local_path), str(remote_path)))

    def add_local_python_source(self, *modules, copy: bool = False) -> "Image":
        return self._with("add_pysource", list(modules))

    def run_function(self, fn: Callable, gpu=None, volumes=None, secrets=None, **kw) -> "Image":
        """Build-time function execution (llm-serving/sglang_snapshot.py:145-149).
        Runs once, lazily, on first use of this image — in a WORKER process
        with the requested gpu/volumes/secrets (reference semantics: build
        steps execute in a container with those resources, e.g. weight
        downloads onto a volume), not in the client process."""
        return self._with("run_function",
                          (fn, {"gpu": gpu, "volumes": volumes, "secrets": secrets}))

    def pip_install_from_requirements(self, path, **kw) -> "Image":
        return self._with("pip_req", str(path))

    def dockerfile_commands(self, *cmds, **kw) -> "Image":
        return self._with("dockerfile", list(cmds))

    # ---- runtime surface ----

    @contextlib.contextmanager
    def imports(self):
        """Deferred-import block (reference idiom throughout, e.g.
        text_to_image.py:68-75): suppress import errors at module scope; they
        resurface when the function actually runs in a worker."""
        try:
            yield
        except ImportError:
            pass

    def content_hash(self) -> str:
        blob = json.dumps(
            [self.base] + [(k, repr(p)) for k, p in self.layers if k != "run_function"],
            sort_keys=True,
        ).encode()
        return hashlib.sha256(blob).hexdigest()[:16]

    # ---- environment materialization (venv isolation) ----

    @property
    def _pip_layers(self):
        return [p for k, p in self.layers if k in ("pip", "pip_req")]

    def build_venv(self) -> Optional[str]:
        """Materialize the pip layers as a content-hashed venv the workers
        exec from (reference: 02_building_containers/install_flash_attn.py:
        17-24 — an image's installs define the interpreter environment, not
        an importability assertion).

        Offline semantics (this node has no network): the venv is created
        with --system-site-packages and each pip layer installs with
        --no-index, which succeeds for anything the local wheel environment
        already satisfies and FAILS LOUDLY for a requirement it cannot meet —
        the build error the reference would surface at image-build time.
        Returns the venv's python path, or None if there are no pip layers.
        """
        if not self._pip_layers:
            return None
        import subprocess
        import sys

        from .. import config

        root = config.state_dir() / "images" / self.content_hash()
        venv = root / "venv"
        py = venv / "bin" / "python"
        manifest = root / "manifest.json"
        if manifest.exists() and py.exists():
            return str(py)
        root.mkdir(parents=True, exist_ok=True)
        import venv as venv_mod

        # with_pip=False: this distro python ships no ensurepip wheels; the
        # SYSTEM pip installs into the venv via --prefix instead (and, with
        # --no-index, validates that each requirement is already satisfiable
        # from the local wheel environment)
        venv_mod.EnvBuilder(system_site_packages=True, with_pip=False,
                            symlinks=True).create(str(venv))
        installed = []
        for pkgs in self._pip_layers:
            args = list(pkgs) if isinstance(pkgs, list) else ["-r", str(pkgs)]
            r = subprocess.run(
                [sys.executable, "-m", "pip", "install", "--no-index",
                 "--prefix", str(venv), *args],
                capture_output=True, text=True)
            if r.returncode != 0:
                raise RuntimeError(
                    f"image build failed: pip layer {args} cannot be satisfied "
                    f"offline:\n{r.stderr[-2000:]}")
            installed.append(args)
        manifest.write_text(json.dumps({
            "base": self.base, "python": sys.version, "pip": installed,
            "hash": self.content_hash()}))
        return str(py)

    @property
    def build_env(self) -> dict:
        return dict(self._env)

    def build(self):
        """Execute run_function layers (once per process), each in a one-shot
        worker carrying the layer's gpu/volumes/secrets request."""
        for entry in self._build_fns:
            fn, opts = entry if isinstance(entry, tuple) else (entry, {})
            key = f"_built_{id(fn)}"
            if getattr(self, key, False):
                continue
            self._run_build_fn(fn, opts or {})
            setattr(self, key, True)
        return self

    def _run_build_fn(self, fn: Callable, opts: dict) -> None:
        from ..app import App  # lazy: resources must not import app at module load

        app = App(f"image-build-{self.content_hash()}")
        builder = app.function(
            gpu=opts.get("gpu"),
            volumes=opts.get("volumes") or {},
            secrets=opts.get("secrets") or [],
            single_use_containers=True,  # build steps are one-shot: don't
            scaledown_window=1.0,        # leave warm workers behind
        )(fn)
        builder.remote()
