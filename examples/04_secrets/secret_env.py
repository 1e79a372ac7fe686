# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/04_secrets/secret_env.py"]
# ---
# Secrets inject env vars into worker processes from the local keystore
# (or `from_dict` for ad-hoc values).

import os

import modal_examples_amd as modal

app = modal.App("example-secrets")

api_secret = modal.Secret.from_dict({"DEMO_API_KEY": "sk-local-123"})


@app.function(secrets=[api_secret])
def use_key() -> str:
    key = os.environ["DEMO_API_KEY"]
    return f"worker saw key ending ...{key[-3:]}"


@app.local_entrypoint()
def main():
    print(use_key.remote())
    assert "123" in use_key.remote()
