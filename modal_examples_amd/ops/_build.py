"""In-tree hipcc build of the gfx950 kernel extension.

Deliberately NOT torch.utils.cpp_extension.load(): that path hipifies CUDA
sources and JIT-caches under ~/.cache, which does not travel to the GPU box.
Here each .hip TU compiles standalone (no torch headers → fast), bindings.cpp
is the single torch-including TU, and the linked _hip_ops.so lands in-tree so
the gpurun snapshot carries it.
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
BUILD = OPS_DIR / "build"
SO_PATH = OPS_DIR / "_hip_ops.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

HIP_SOURCES = [
    "attention_fwd.hip",
    "attention_fwd32.hip",
    "attention_decode.hip",
    "norms.hip",
    "elementwise.hip",
    "adamw.hip",
    "sampling.hip",
    "conv3x3.hip",
]


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"build failed: {' '.join(cmd)}\n--- stdout ---\n{r.stdout[-4000:]}"
            f"\n--- stderr ---\n{r.stderr[-8000:]}"
        )
    return r


def _mtime(p: Path) -> float:
    return p.stat().st_mtime if p.exists() else 0.0


def _torch_flags():
    import torch
    from torch.utils import cpp_extension as ce

    includes = [f"-I{p}" for p in ce.include_paths("cuda")]
    includes.append(f"-I{sys.prefix}/include/python{sys.version_info.major}.{sys.version_info.minor}")
    import sysconfig

    includes.append(f"-I{sysconfig.get_paths()['include']}")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defines = [
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ]
    lib_dirs = [f"-L{p}" for p in ce.library_paths("cuda")]
    libs = ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
            "-ltorch_python", "-lamdhip64"]
    return includes, defines, lib_dirs, libs


def build(verbose: bool = False, force: bool = False) -> Path:
    """Compile every HIP TU for gfx950 and link the torch extension in-tree."""
    BUILD.mkdir(exist_ok=True)
    hipcc = os.environ.get("HIPCC", "hipcc")
    common = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
              f"-I{CSRC}"]
    objs = []
    hdr_m = _mtime(CSRC / "common.h")
    relink = force or not SO_PATH.exists()
    for src in HIP_SOURCES:
        sp = CSRC / src
        op = BUILD / (src.replace(".hip", ".o"))
        if force or _mtime(op) < max(_mtime(sp), hdr_m):
            if verbose:
                print(f"[ops] hipcc -c {src}", flush=True)
            _run([hipcc, *common, "-c", str(sp), "-o", str(op)])
            relink = True
        objs.append(str(op))

    includes, defines, lib_dirs, libs = _torch_flags()
    bp = CSRC / "bindings.cpp"
    bo = BUILD / "bindings.o"
    if force or _mtime(bo) < max(_mtime(bp), hdr_m):
        if verbose:
            print("[ops] hipcc -c bindings.cpp (torch TU)", flush=True)
        _run([hipcc, *common, *includes, *defines, "-fvisibility=hidden",
              "-c", str(bp), "-o", str(bo)])
        relink = True
    objs.append(str(bo))

    if relink:
        if verbose:
            print("[ops] linking _hip_ops.so", flush=True)
        _run([hipcc, "-shared", "-fPIC", *objs, *lib_dirs, *libs,
              "-o", str(SO_PATH)])
    return SO_PATH


_ext = None
_ext_err = None


def get_ext(required: bool = False):
    """Import the built extension; build it if missing. Returns None on CPU-only
    hosts unless required=True."""
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    try:
        import torch  # noqa: F401 — loads libtorch into the process first

        if not SO_PATH.exists():
            build()
        import importlib.util

        spec = importlib.util.spec_from_file_location("_hip_ops", SO_PATH)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        return _ext
    except Exception as e:  # noqa: BLE001
        _ext_err = e
        if required:
            raise RuntimeError(
                f"gfx950 kernel extension unavailable: {e}. On a GPU box this "
                "is fatal — the HIP path must run, not a torch fallback."
            ) from e
        return None


if __name__ == "__main__":
    p = build(verbose=True, force="--force" in sys.argv)
    print(f"built {p}")
