# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/bucket_mount_loras.py"]
# ---
# # Serving LoRA adapters from a cloud bucket (cloud_bucket_mount_loras role)
#
# Community LoRA weights live in an object store; inference containers mount
# the bucket READ-ONLY and hot-load adapters by name
# (10_integrations/cloud_bucket_mount_loras.py).  Here: adapters are
# published to the local S3 endpoint, a GPU class mounts the prefix and
# applies a chosen adapter to its model; the self-check verifies each
# adapter steers the output its own way.

import modal_examples_amd as modal

app = modal.App("example-bucket-loras")

BUCKET = "lora-hub"
loras = modal.CloudBucketMount(BUCKET, key_prefix="sdxl-adapters",
                               read_only=True)


def build_net(torch):
    import torch.nn as nn

    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(64, 128), nn.SiLU(), nn.Linear(128, 64))


@app.cls(gpu="mi355x", volumes={"/mnt/loras": loras})
class StyledModel:
    @modal.enter()
    def load(self):
        import torch

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        self.net = build_net(torch).to(self.device).eval()
        self.active = None

    @modal.method()
    def list_adapters(self) -> list:
        from pathlib import Path

        return sorted(p.stem for p in Path("/mnt/loras").glob("*.pt"))

    @modal.method()
    def stylize(self, adapter: str) -> float:
        """Apply the named adapter from the bucket; return an output probe."""
        import torch

        from modal_examples_amd.train.lora import apply_lora, load_lora_state

        net = build_net(torch).to(self.device).eval()
        apply_lora(net, rank=4, targets=("0", "2"))
        state = torch.load(f"/mnt/loras/{adapter}.pt",
                           map_location=self.device)
        load_lora_state(net, state)
        torch.manual_seed(7)
        x = torch.randn(8, 64, device=self.device)
        with torch.no_grad():
            return float(net(x).sum())


@app.local_entrypoint()
def main():
    import io

    import torch

    from modal_examples_amd.resources.s3local import S3Client, start_s3_server
    from modal_examples_amd.train.lora import apply_lora, lora_state_dict

    # publish two adapters to the bucket (the community-hub producer)
    c = S3Client(start_s3_server())
    for name, seed in (("watercolor", 1), ("neon", 2)):
        net = build_net(torch)  # (re-seeds to 0 internally)
        # nn.Sequential children are named "0"/"2" — target them explicitly
        apply_lora(net, rank=4, targets=("0", "2"))
        torch.manual_seed(seed)  # per-adapter weights AFTER the base build
        for p in net.parameters():  # give the adapter non-zero weights
            if p.requires_grad and p.dim() == 2:
                torch.nn.init.normal_(p, std=0.2)
        buf = io.BytesIO()
        torch.save(lora_state_dict(net), buf)
        c.put(BUCKET, f"sdxl-adapters/{name}.pt", buf.getvalue())

    m = StyledModel()
    found = m.list_adapters.remote()
    assert found == ["neon", "watercolor"], found
    a = m.stylize.remote("watercolor")
    b = m.stylize.remote("neon")
    a2 = m.stylize.remote("watercolor")
    assert abs(a - a2) < 1e-3 and abs(a - b) > 1e-3, (a, b, a2)
    print(f"adapters {found}: watercolor probe {a:.3f}, neon probe {b:.3f}")
