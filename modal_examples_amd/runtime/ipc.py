"""Serialization + message shapes for client↔worker IPC.

All payloads crossing a process boundary are cloudpickle bytes: the examples the
reference corpus models (e.g. 01_getting_started/hello_world.py) define functions
in ``__main__``/test modules, which plain pickle cannot ship to a spawned child.
"""
from __future__ import annotations

import pickle
from dataclasses import dataclass, field
from typing import Any, Optional

import cloudpickle


def dumps(obj: Any) -> bytes:
    return cloudpickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)


def loads(blob: bytes) -> Any:
    return pickle.loads(blob)


@dataclass
class ServiceSpec:
    """Everything a worker needs to host one Function (or one Cls instance)."""

    app_name: str
    name: str
    target_blob: bytes  # cloudpickled callable OR (class, params dict)
    is_cls: bool = False
    cls_params: dict = field(default_factory=dict)
    max_inputs: int = 1  # @modal.concurrent input slots per worker
    gpu_devices: tuple = ()  # device indices assigned to this worker
    env: dict = field(default_factory=dict)  # secrets etc, injected pre-import
    volumes: dict = field(default_factory=dict)  # mount_path -> volume name
    is_generator: bool = False
    batched: bool = False
    enable_memory_snapshot: bool = False
    enable_gpu_snapshot: bool = False


# worker → supervisor message kinds
READY = "ready"  # enter hooks done, worker accepting inputs
RESULT = "result"  # (call_id, value_blob)
ERROR = "error"  # (call_id, exc_blob, traceback_str)
YIELD = "yield"  # (call_id, item_blob)  — generator item
GEN_END = "gen_end"  # (call_id,) — generator exhausted
EXITED = "exited"  # worker ran @exit hooks and is quitting
LOG = "log"


@dataclass
class WorkerMsg:
    worker_id: int
    kind: str
    call_id: Optional[str] = None
    payload: Optional[bytes] = None
    text: str = ""


# supervisor → worker task kinds
T_CALL = "call"  # one invocation: (call_id, method_name, args_blob)
T_BATCH = "batch"  # batched invocation: (call_ids, args_blobs)
T_SHUTDOWN = "shutdown"
