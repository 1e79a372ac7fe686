"""MIOpen find-db caching: the compile-artifact-cache trick, MI355X edition.

The reference caches torch.compile/inductor/triton artifacts on Volumes so
warm boots skip recompilation (stable_diffusion/flux.py:70-75,115-122).  The
MIOpen analog: `torch.backends.cudnn.benchmark` exhaustively times conv algos
per shape (minutes for the SDXL VAE at 1024px) and stores results in MIOpen's
user find-db under $HOME — which is empty on every fresh box.  This module
saves/restores that db from a directory that travels with the repo (or a
Volume), turning a ~3-minute first-generate into seconds.
"""
from __future__ import annotations

import os
import shutil
from pathlib import Path

DEFAULT_CACHE = Path(__file__).resolve().parent / "miopen_udb"


def _user_db_dir() -> Path:
    base = os.environ.get("MIOPEN_USER_DB_PATH")
    if base:
        return Path(base)
    return Path(os.path.expanduser("~")) / ".config" / "miopen"


def restore(cache_dir: Path = DEFAULT_CACHE) -> int:
    """Copy a cached find-db into MIOpen's user db location. Returns files
    restored (0 = no cache yet).  Multi-rank safe: writes go to a temp file
    then os.replace (atomic), so 8 concurrent ranks cannot corrupt the db."""
    cache_dir = Path(cache_dir)
    if not cache_dir.is_dir():
        return 0
    dst = _user_db_dir()
    dst.mkdir(parents=True, exist_ok=True)
    n = 0
    for f in cache_dir.iterdir():
        if f.is_file():
            target = dst / f.name
            if not target.exists() or target.stat().st_size != f.stat().st_size:
                tmp = dst / f"{f.name}.tmp{os.getpid()}"
                shutil.copy2(f, tmp)
                os.replace(tmp, target)
            n += 1
    return n


def capture(cache_dir: Path = DEFAULT_CACHE) -> int:
    """Copy MIOpen's user find-db into the repo-tracked cache dir."""
    src = _user_db_dir()
    cache_dir = Path(cache_dir)
    cache_dir.mkdir(parents=True, exist_ok=True)
    n = 0
    if src.is_dir():
        for f in src.iterdir():
            if f.is_file() and f.suffix in (".fdb", ".udb", ".txt", ""):
                shutil.copy2(f, cache_dir / f.name)
                n += 1
    return n
