"""Worker-process main loop.

One worker process ≈ one Modal "container" (reference semantics at SURVEY.md §2.2:
container≈worker-process lifecycle).  A worker:

  1. pins its assigned GPUs via ``HIP_VISIBLE_DEVICES`` *before* any torch import,
  2. unpickles its service target (function or class),
  3. runs ``@enter`` lifecycle hooks (snap=True phase first, then snap=False —
     mirroring the phase split at reference 06_gpu_and_ml/gpu_snapshot.py:41-53),
  4. serves tasks from its task queue with ``max_inputs`` concurrent slots
     (``@modal.concurrent``, reference 06_gpu_and_ml/llm-serving/sglang_snapshot.py:260),
  5. on shutdown runs ``@exit`` hooks (reference lifecycle decorator counts: §1 L5).
"""
from __future__ import annotations

import os
from pathlib import Path
import queue as _queue
import threading
import traceback
from concurrent.futures import ThreadPoolExecutor

from . import ipc

_call_ctx = threading.local()


def current_call_id():
    """The FunctionCall id of the invocation running on THIS thread (None
    outside a call) — backs modal.current_function_call_id()."""
    return getattr(_call_ctx, "call_id", None)


def _bind_read_only(pairs: list) -> bool:
    """Enforce read-only mounts at the filesystem level: unshare this worker's
    mount namespace (mounts stay private to the process) and bind-mount each
    volume directory read-only at its mount path.  Returns False if the kernel
    refuses (no CAP_SYS_ADMIN) — callers fall back to a plain symlink with
    API-level enforcement only.  Must run before any threads start:
    unshare(CLONE_NEWNS) fails EINVAL in a multithreaded process."""
    import ctypes

    try:
        libc = ctypes.CDLL("libc.so.6", use_errno=True)
        CLONE_NEWNS = 0x00020000
        MS_RDONLY, MS_BIND, MS_REC, MS_REMOUNT, MS_PRIVATE = 1, 0x1000, 0x4000, 32, 0x40000
        if libc.unshare(CLONE_NEWNS) != 0:
            return False
        if libc.mount(b"none", b"/", None, MS_REC | MS_PRIVATE, None) != 0:
            return False
        for mount, target in pairs:
            os.makedirs(mount, exist_ok=True)
            m, t = str(mount).encode(), str(target).encode()
            if libc.mount(t, m, None, MS_BIND, None) != 0:
                return False
            if libc.mount(b"none", m, None, MS_BIND | MS_REMOUNT | MS_RDONLY, None) != 0:
                return False
        return True
    except OSError:
        return False


_bucket_syncs = []  # (bucket, prefix, local_dir) pending write-back


def _mount_bucket(mount: str, name: str, ro: bool):
    """s3:<bucket>:<prefix> mount: prefix-sync download into a private dir
    via the local S3 endpoint; non-ro mounts write back at worker exit.
    Returns the private dir for ro mounts (caller bind-mounts it read-only)
    or None when the mount was placed here."""
    import tempfile

    from ..resources.s3local import S3Client

    _, bucket, prefix = name.split(":", 2)
    endpoint = os.environ.get("MODAL_AMD_S3_ENDPOINT", "")
    local = Path(tempfile.mkdtemp(prefix=f"s3mount-{bucket}-"))
    if endpoint:
        try:
            S3Client(endpoint).sync_down(bucket, prefix, local)
        except Exception:
            pass  # empty bucket / server racing: mount starts empty
    else:
        # no endpoint (direct in-process use): fall back to the server-side dir
        from .. import config

        src = config.state_dir() / "buckets" / bucket / prefix
        if src.exists():
            import shutil

            shutil.copytree(src, local, dirs_exist_ok=True)
    if ro:
        return local  # caller adds (mount, local) to the RO bind-mount set
    try:
        if os.path.islink(mount):
            os.unlink(mount)
        if not os.path.exists(mount):
            parent = os.path.dirname(str(mount).rstrip("/"))
            if parent and not os.path.exists(parent):
                os.makedirs(parent, exist_ok=True)
            os.symlink(local, mount)
    except OSError:
        pass
    _bucket_syncs.append((bucket, prefix, local))
    return None


def _writeback_buckets() -> None:
    from ..resources.s3local import S3Client

    endpoint = os.environ.get("MODAL_AMD_S3_ENDPOINT", "")
    for bucket, prefix, local in _bucket_syncs:
        try:
            if endpoint:
                S3Client(endpoint).sync_up(bucket, prefix, local)
            else:
                from .. import config

                import shutil

                dst = config.state_dir() / "buckets" / bucket / prefix
                shutil.copytree(local, dst, dirs_exist_ok=True)
        except Exception:
            pass


def _mount_volumes(volumes: dict) -> None:
    """Symlink mount paths to the shared volume directories (the worker runs
    as root on this node — mirrors containers mounting at /cache etc.).
    Names carrying an ``ro:`` prefix (Volume.read_only()) get a read-only
    bind mount in a private mount namespace instead of a symlink."""
    from .. import config

    ro_pairs = []
    for mount, name in (volumes or {}).items():
        ro = name.startswith("ro:")
        if ro:
            name = name[3:]
        if name.startswith("s3:"):
            ro_target = _mount_bucket(mount, name, ro)
            if ro_target is not None:
                ro_pairs.append((mount, ro_target))
            continue
        if name.startswith("bucket:"):
            target = config.state_dir() / "buckets" / name.split(":", 1)[1]
        else:
            target = config.state_dir() / "volumes" / name
        target.mkdir(parents=True, exist_ok=True)
        try:
            if os.path.islink(mount):
                if os.path.realpath(mount) == os.path.realpath(target):
                    continue
                os.unlink(mount)  # stale link into another state dir
            if ro:
                ro_pairs.append((mount, target))  # bind over dir is fine
                continue
            if os.path.exists(mount):
                continue
            parent = os.path.dirname(mount.rstrip("/"))
            if parent and not os.path.exists(parent):
                os.makedirs(parent, exist_ok=True)
            os.symlink(target, mount)
        except OSError:
            pass  # unmountable path: functions can still use volume.path
    if ro_pairs:
        if _bind_read_only(ro_pairs):
            # user code can check whether raw-write protection is active
            os.environ["MODAL_AMD_RO_ENFORCED"] = "1"
        else:
            for mount, target in ro_pairs:  # fallback: API-level only (the
                try:                        # kernel refused CAP_SYS_ADMIN)
                    if not os.path.exists(mount):
                        os.symlink(target, mount)
                except OSError:
                    pass


def _apply_env(spec_env: dict, gpu_devices: tuple) -> None:
    if gpu_devices:
        vis = ",".join(str(d) for d in gpu_devices)
        os.environ["HIP_VISIBLE_DEVICES"] = vis
        os.environ["CUDA_VISIBLE_DEVICES"] = vis
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    for k, v in spec_env.items():
        os.environ[k] = str(v)


def _resolve_target(spec: "ipc.ServiceSpec"):
    """Returns (callable_map, exit_hooks). callable_map: method_name -> callable."""
    target = ipc.loads(spec.target_blob)
    if not spec.is_cls:
        return {"": target}, []

    cls = target
    obj = cls()
    # Bind modal.parameter() fields (reference: stable_diffusion/flux.py:126-128)
    from ..app import _iter_parameters

    for pname, default in _iter_parameters(cls):
        setattr(obj, pname, spec.cls_params.get(pname, default))

    # Lifecycle hooks, snapshot phase first (gpu_snapshot.py:41-53 ordering)
    enter_hooks = []
    exit_hooks = []
    methods = {}
    for name in dir(cls):
        fn = getattr(cls, name, None)
        flags = getattr(fn, "_modal_flags", None)
        if not flags:
            continue
        bound = getattr(obj, name)
        if flags.get("enter"):
            enter_hooks.append((bool(flags.get("snap")), bound))
        if flags.get("exit"):
            exit_hooks.append(bound)
        if flags.get("method") or flags.get("web") or flags.get("batched"):
            methods[name] = bound
    for snap_phase in (True, False):
        for is_snap, hook in enter_hooks:
            if is_snap == snap_phase:
                hook()
    methods.setdefault("", None)
    return methods, exit_hooks


def worker_main(worker_id: int, spec_blob: bytes, task_q, result_q) -> None:
    spec: ipc.ServiceSpec = ipc.loads(spec_blob)
    _apply_env(spec.env, spec.gpu_devices)
    _mount_volumes(spec.volumes)
    os.environ["MODAL_TASK_ID"] = f"ta-local-{spec.name}-{worker_id}"

    def post(kind, call_id=None, payload=None, text=""):
        result_q.put(ipc.WorkerMsg(worker_id, kind, call_id, payload, text))

    try:
        methods, exit_hooks = _resolve_target(spec)
    except BaseException as e:  # noqa: BLE001
        post(ipc.ERROR, None, ipc.dumps(e), traceback.format_exc())
        return
    post(ipc.READY)

    import inspect

    # `modal shell` into THIS running container: exec code in a persistent
    # namespace that can see the live service (loaded models, caches, ...).
    debug_ns = {"methods": methods, "spec": spec, "os": os}
    debug_ns["obj"] = getattr(  # the live class instance, for @app.cls services
        next((m for m in methods.values() if m is not None), None), "__self__", None)

    def _debug_exec(src: str):
        import io
        from contextlib import redirect_stdout

        buf = io.StringIO()
        with redirect_stdout(buf):
            try:
                value = eval(compile(src, "<shell>", "eval"), debug_ns)
                if value is not None:
                    print(repr(value))
            except SyntaxError:
                exec(compile(src, "<shell>", "exec"), debug_ns)
        return buf.getvalue()

    def run_one(call_id: str, method_name: str, args_blob: bytes) -> None:
        _call_ctx.call_id = call_id
        try:
            args, kwargs = ipc.loads(args_blob)
            if method_name == "__debug_exec__":
                post(ipc.RESULT, call_id, ipc.dumps(_debug_exec(args[0])))
                return
            fn = methods.get(method_name)
            if fn is None:
                raise RuntimeError(f"no method {method_name!r} on service {spec.name}")
            raw = getattr(fn, "_modal_raw", fn)
            if inspect.isgeneratorfunction(raw):
                for item in fn(*args, **kwargs):
                    post(ipc.YIELD, call_id, ipc.dumps(item))
                post(ipc.GEN_END, call_id)
            else:
                out = fn(*args, **kwargs)
                post(ipc.RESULT, call_id, ipc.dumps(out))
        except BaseException as e:  # noqa: BLE001
            try:
                blob = ipc.dumps(e)
            except Exception:
                blob = ipc.dumps(RuntimeError(repr(e)))
            post(ipc.ERROR, call_id, blob, traceback.format_exc())
        finally:
            _call_ctx.call_id = None

    def run_batch(call_ids, method_name: str, args_blobs) -> None:
        """@modal.batched: collect single-input calls into one list-shaped call.

        The wrapped function receives lists and must return a parallel list
        (reference: 03_scaling_out/dynamic_batching.py:29-45).
        """
        try:
            unpacked = [ipc.loads(b) for b in args_blobs]
            fn = methods.get(method_name)
            # each call's args: (args_tuple, kwargs). Batched fns take positional
            # lists: transpose the per-call positional args into per-arg lists.
            nargs = max(len(a) for a, _ in unpacked)
            cols = [[a[i] for a, _ in unpacked] for i in range(nargs)]
            outs = fn(*cols)
            if len(outs) != len(call_ids):
                raise RuntimeError(
                    f"batched function returned {len(outs)} outputs for {len(call_ids)} inputs"
                )
            for cid, out in zip(call_ids, outs):
                post(ipc.RESULT, cid, ipc.dumps(out))
        except BaseException as e:  # noqa: BLE001
            tb = traceback.format_exc()
            for cid in call_ids:
                post(ipc.ERROR, cid, ipc.dumps(e), tb)

    pool = ThreadPoolExecutor(max_workers=max(1, spec.max_inputs))
    while True:
        try:
            task = task_q.get(timeout=3600.0)
        except _queue.Empty:
            continue
        kind = task[0]
        if kind == ipc.T_SHUTDOWN:
            break
        if kind == ipc.T_CALL:
            _, call_id, method_name, args_blob = task
            if spec.max_inputs > 1:
                pool.submit(run_one, call_id, method_name, args_blob)
            else:
                run_one(call_id, method_name, args_blob)
        elif kind == ipc.T_BATCH:
            _, call_ids, method_name, args_blobs = task
            run_batch(call_ids, method_name, args_blobs)

    pool.shutdown(wait=True)
    _writeback_buckets()
    for hook in exit_hooks:
        try:
            hook()
        except Exception:  # noqa: BLE001
            traceback.print_exc()
    post(ipc.EXITED)
