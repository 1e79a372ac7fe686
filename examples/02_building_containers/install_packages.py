# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/02_building_containers/install_packages.py"]
# ---
# # Installing packages into an isolated image environment
#
# The install_flash_attn.py role: an Image's pip layers define the worker's
# interpreter environment.  Locally the layers materialize a content-hashed
# venv (offline resolution against the node's wheel set) and the worker
# EXECS that venv's python — then the function proves the install by running
# the package's hot path (here: the gfx950 flash-attention op, the
# flash_attn shape-check idiom at install_flash_attn.py:46).

import modal_examples_amd as modal

app = modal.App("example-install-packages")

image = (
    modal.Image.debian_slim(python_version="3.10")
    .uv_pip_install("numpy", "einops")
    .env({"MIOPEN_FIND_MODE": "FAST"})
)


@app.function(gpu="mi355x", image=image)
def check_install() -> dict:
    import sys

    import einops
    import torch

    from modal_examples_amd.ops import functional as OF

    q = torch.randn(1, 4, 64, 64, dtype=torch.bfloat16)
    if torch.cuda.is_available():
        q = q.cuda()
    o = OF.attention(q, q, q)
    rearranged = einops.rearrange(o, "b h s d -> b s (h d)")
    return {
        "python_prefix": sys.prefix,
        "isolated": sys.prefix != sys.base_prefix,
        "attn_shape": list(rearranged.shape),
    }


@app.local_entrypoint()
def main():
    out = check_install.remote()
    assert out["isolated"], out  # worker ran the image's venv interpreter
    assert out["attn_shape"] == [1, 64, 256]
    print(f"venv: {out['python_prefix']}")
    print(f"attention output through the image env: {out['attn_shape']}")
