"""Volume: shared directory with commit/reload semantics.

Reference usage: 118 ``Volume.from_name`` sites; commit/reload at
06_gpu_and_ml/dreambooth/diffusers_lora_finetune.py:343,367.  Locally a volume
is a directory under the state dir, shared by all worker processes (same host),
so ``commit``/``reload`` are consistency barriers: commit fsyncs, reload is a
no-op that revalidates existence.  Workers mount volumes by symlinking the
mount path to the volume directory (they run as root on this node).
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Optional

from .. import config
from ..exception import NotFoundError


class Volume:
    def __init__(self, name: str, _create: bool = True, version: int = 2):
        self.name = name
        self.version = version
        p = self.path
        if _create:
            p.mkdir(parents=True, exist_ok=True)
        elif not p.exists():
            raise NotFoundError(f"volume {name!r} does not exist")

    @property
    def path(self) -> Path:
        return config.state_dir() / "volumes" / self.name

    @staticmethod
    def from_name(name: str, create_if_missing: bool = False, version: int = 2) -> "Volume":
        return Volume(name, _create=create_if_missing, version=version)

    @staticmethod
    def ephemeral():
        import contextlib
        import uuid

        @contextlib.contextmanager
        def ctx():
            v = Volume(f"ephemeral-{uuid.uuid4().hex[:8]}")
            try:
                yield v
            finally:
                import shutil

                shutil.rmtree(v.path, ignore_errors=True)

        return ctx()

    def read_only(self) -> "Volume":
        """A read-only view for mounting (reference restricted-volume usage,
        08_advanced).  Write APIs raise ``InvalidError``; workers additionally
        enforce it at the filesystem level by bind-mounting the volume
        read-only inside a private mount namespace (runtime/worker.py)."""
        return _ReadOnlyVolume(self)

    def commit(self):
        """Flush writes so other workers observe them (fsync the tree)."""
        d = os.open(self.path, os.O_RDONLY)
        try:
            os.fsync(d)
        finally:
            os.close(d)

    def reload(self):
        if not self.path.exists():
            raise NotFoundError(f"volume {self.name!r} vanished")

    def listdir(self, path: str = "/", recursive: bool = False):
        base = self.path / path.lstrip("/")
        if recursive:
            out = []
            for root, _dirs, files in os.walk(base):
                rel = Path(root).relative_to(self.path)
                out.extend(str(rel / f) for f in files)
            return out
        return [p.name for p in base.iterdir()]

    def iterdir(self, path: str = "/"):
        yield from self.listdir(path)

    def read_file(self, path: str) -> bytes:
        return (self.path / path.lstrip("/")).read_bytes()

    def remove_file(self, path: str, recursive: bool = False):
        p = self.path / path.lstrip("/")
        if recursive:
            import shutil

            shutil.rmtree(p)
        else:
            p.unlink()

    @staticmethod
    def delete(name: str):
        import shutil

        shutil.rmtree(config.state_dir() / "volumes" / name, ignore_errors=True)

    def batch_upload(self):
        vol = self

        class _Batch:
            def __enter__(self):
                return self

            def __exit__(self, *a):
                vol.commit()
                return False

            def put_file(self, local, remote):
                import shutil

                dst = vol.path / str(remote).lstrip("/")
                dst.parent.mkdir(parents=True, exist_ok=True)
                shutil.copy2(local, dst)

            def put_directory(self, local, remote):
                import shutil

                dst = vol.path / str(remote).lstrip("/")
                shutil.copytree(local, dst, dirs_exist_ok=True)

        return _Batch()


class _ReadOnlyVolume(Volume):
    """Read-only view over a Volume.  ``name`` carries an ``ro:`` prefix that
    the worker mount logic interprets as "bind-mount read-only"."""

    def __init__(self, base: Volume):  # noqa: super-init-not-called (view)
        self._base = base
        self.version = base.version

    @property
    def name(self) -> str:  # type: ignore[override]
        return f"ro:{self._base.name}"

    @property
    def path(self) -> Path:
        return self._base.path

    def read_only(self) -> "Volume":
        return self

    def _refuse(self):
        from ..exception import InvalidError

        raise InvalidError(f"volume {self._base.name!r} is mounted read-only")

    def commit(self):
        self._refuse()

    def remove_file(self, path: str, recursive: bool = False):
        self._refuse()

    def batch_upload(self):
        self._refuse()


class CloudBucketMount:
    """S3-backed bucket mount (10_integrations/s3_bucket_mount.py:63-80).

    Real bucket semantics on a no-network node: objects live behind a LOCAL
    S3-compatible REST endpoint (resources/s3local.py); mounting a bucket in
    a worker prefix-DOWNLOADS the objects into a private per-worker directory
    over HTTP (not a shared-dir symlink) and, unless read_only, writes
    changes back to the endpoint at worker shutdown."""

    def __init__(self, bucket_name: str, secret=None, read_only: bool = False,
                 key_prefix: Optional[str] = None, **kw):
        self.bucket_name = bucket_name
        self.read_only = read_only
        self.key_prefix = (key_prefix or "").strip("/")
        # direct server-side path (client-side seeding/tests); workers see a
        # synced copy, never this directory
        self.path = config.state_dir() / "buckets" / bucket_name
        self.path.mkdir(parents=True, exist_ok=True)

    @property
    def name(self) -> str:
        base = f"s3:{self.bucket_name}:{self.key_prefix}"
        return f"ro:{base}" if self.read_only else base
