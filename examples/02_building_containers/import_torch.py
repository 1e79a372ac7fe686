# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/02_building_containers/import_torch.py"]
# ---
# # Environment sanity: ROCm torch + the gfx950 kernel extension
#
# The install_cuda/import_torch analog: verify the worker environment has
# PyTorch-ROCm, report the GPU, and check the in-tree HIP extension loads.

import modal_examples_amd as modal

app = modal.App("example-import-torch")

image = modal.Image.debian_slim().env({"MIOPEN_FIND_MODE": "FAST"})


@app.function(gpu="mi355x", image=image)
def check_env() -> dict:
    import torch

    info = {
        "torch": torch.__version__,
        "rocm": torch.version.hip,
        "cuda_available": torch.cuda.is_available(),
    }
    if torch.cuda.is_available():
        info["device"] = torch.cuda.get_device_name(0)
        info["hbm_gb"] = round(torch.cuda.mem_get_info()[1] / 1e9, 1)
        from modal_examples_amd.ops._build import get_ext

        info["hip_ext"] = get_ext(required=True) is not None
    return info


@app.local_entrypoint()
def main():
    info = check_env.remote()
    for k, v in info.items():
        print(f"{k}: {v}")
    assert info["rocm"] is not None
