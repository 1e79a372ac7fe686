"""modal_examples_amd — an MI355X-native serverless-GPU runtime with the API
surface of the Modal examples corpus (modal-labs/modal-examples).

Import style mirrors the reference examples::

    import modal_examples_amd as modal

    app = modal.App("example")

    @app.function(gpu="mi355x")
    def f(x): ...

The runtime schedules decorated functions onto a local pool of MI355X GPUs
(one worker process per container, ``HIP_VISIBLE_DEVICES`` pinning), with the
compute path built on PyTorch-ROCm + hand-written gfx950 HIP kernels (ops/) and
RCCL over xGMI for collectives (parallel/).
"""
from . import experimental  # noqa: F401
from .app import (  # noqa: F401
    App,
    Cls,
    Cron,
    Function,
    FunctionCall,
    Period,
    Retries,
    Stub,
    batched,
    concurrent,
    enable_output,
    enter,
    exit,
    gather,
    method,
    parameter,
)
from .exception import (  # noqa: F401
    Error,
    FunctionTimeoutError,
    InvalidError,
    NotFoundError,
    RemoteError,
)
from .resources.dict_queue import Dict, Queue  # noqa: F401
from .resources.image import Image  # noqa: F401
from .resources.secret import Secret  # noqa: F401
from .resources.volume import CloudBucketMount, Volume  # noqa: F401

# modal-API alias: NetworkFileSystem is the older shared-writable-FS object;
# Volume covers its semantics here (same commit/reload-free shared dir)
NetworkFileSystem = Volume
from .runtime.sandbox import ContainerProcess, Probe, Sandbox, Tunnel  # noqa: F401
from .web.ingress import (  # noqa: F401
    asgi_app,
    fastapi_endpoint,
    forward,
    web_endpoint,
    web_server,
    wsgi_app,
)

__version__ = "0.1.0"


def is_local() -> bool:
    """True in the client process, False inside a worker "container"
    (reference idiom: guard client-only imports/config, e.g.
    llm-serving/openai_compatible/load_test.py:7)."""
    import os

    return not os.environ.get("MODAL_TASK_ID")


def current_function_call_id():
    """Id of the currently-executing invocation (inside a worker; reference:
    09_job_queues/pipeline_orchestration.py:111).  None in the client."""
    from .runtime.worker import current_call_id

    return current_call_id()


# `modal.functions.gather` style access
from . import app as _app_mod


class functions:  # noqa: N801 — mirrors modal.functions module
    gather = staticmethod(_app_mod.gather)
    FunctionCall = _app_mod.FunctionCall


class exception:  # noqa: N801 — mirrors modal.exception module
    from .exception import (
        Error,
        ExecutionError,
        FunctionTimeoutError,
        InvalidError,
        NotFoundError,
        OutputExpiredError,
        RemoteError,
        SandboxTimeoutError,
    )
