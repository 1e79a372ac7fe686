"""Secrets: env-var injection from a local keystore.

Reference: 67 ``Secret.from_name`` sites (e.g. 04_secrets/db_to_sheet.py).
Locally secrets live in ``<state>/secrets.json`` (name → {KEY: value}) or are
taken from the caller's environment; they are injected into worker-process
environments before user code imports.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

from .. import config
from ..exception import NotFoundError


def _keystore() -> Dict[str, Dict[str, str]]:
    p = config.state_dir() / "secrets.json"
    if p.exists():
        return json.loads(p.read_text())
    return {}


def _save_keystore(data):
    (config.state_dir() / "secrets.json").write_text(json.dumps(data, indent=1))


class Secret:
    def __init__(self, env: Dict[str, str], name: str = ""):
        self.env = dict(env)
        self.name = name

    @staticmethod
    def from_name(name: str, required_keys: Optional[List[str]] = None,
                  environment_name: Optional[str] = None) -> "Secret":
        ks = _keystore()
        env = ks.get(name)
        if env is None:
            # fall back: pull required keys from the local environment
            env = {}
            for k in required_keys or []:
                if k in os.environ:
                    env[k] = os.environ[k]
            if required_keys and len(env) != len(required_keys):
                missing = [k for k in required_keys if k not in env]
                raise NotFoundError(
                    f"secret {name!r} not in local keystore and env lacks {missing}"
                )
        if required_keys:
            missing = [k for k in required_keys if k not in env]
            if missing:
                raise NotFoundError(f"secret {name!r} missing required keys {missing}")
        return Secret(env, name)

    @staticmethod
    def from_dict(env: Dict[str, str]) -> "Secret":
        return Secret(env)

    @staticmethod
    def from_local_environ(env_keys: List[str]) -> "Secret":
        return Secret({k: os.environ[k] for k in env_keys if k in os.environ})

    @staticmethod
    def create(name: str, env: Dict[str, str]):
        ks = _keystore()
        ks[name] = dict(env)
        _save_keystore(ks)
        return Secret(env, name)
