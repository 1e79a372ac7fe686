# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/protein_folding.py"]
# ---
# # Protein structure refinement fan-out
#
# The ESMFold/Boltz shape: a batch of sequences, one GPU fold per sequence
# via `.map`, structures to a Volume.  The "fold" here is a physics toy —
# gradient descent of 3D coordinates against a synthetic contact-map energy —
# standing in for the real structure module (same fan-out, same I/O).

import modal_examples_amd as modal

app = modal.App("example-protein-folding")

structures = modal.Volume.from_name("folded-structures", create_if_missing=True)

AA = "ACDEFGHIKLMNPQRSTVWY"


@app.function(gpu="mi355x", timeout=600)
def fold(seq: str) -> dict:
    import hashlib

    import torch

    device = "cuda" if torch.cuda.is_available() else "cpu"
    n = len(seq)
    seed = int.from_bytes(hashlib.sha256(seq.encode()).digest()[:4], "little")
    g = torch.Generator().manual_seed(seed)
    # synthetic target contact map from the sequence
    target = (torch.rand(n, n, generator=g) < 0.08).float()
    target = ((target + target.T) > 0).float().to(device)
    coords = torch.randn(n, 3, generator=g).to(device).requires_grad_(True)
    opt = torch.optim.Adam([coords], lr=0.05)
    for _ in range(150):
        d = torch.cdist(coords, coords)
        contact_e = (target * (d - 3.8) ** 2).mean()
        chain_e = ((d.diagonal(1) - 3.8) ** 2).mean()
        clash_e = torch.relu(3.0 - d + torch.eye(n, device=device) * 10).mean()
        loss = contact_e + chain_e + clash_e
        opt.zero_grad()
        loss.backward()
        opt.step()
    out = coords.detach().cpu()
    name = hashlib.md5(seq.encode()).hexdigest()[:8]
    path = structures.path / f"{name}.xyz"
    with open(path, "w") as f:
        f.write(f"{n}\nfolded {seq[:20]}\n")
        for i, (x, y, z) in enumerate(out.tolist()):
            f.write(f"{seq[i]} {x:.3f} {y:.3f} {z:.3f}\n")
    structures.commit()
    return {"seq_len": n, "final_energy": round(float(loss), 4), "file": path.name}


@app.local_entrypoint()
def main(n_seqs: int = 4):
    import random

    rng = random.Random(0)
    seqs = ["".join(rng.choice(AA) for _ in range(rng.randint(24, 48)))
            for _ in range(n_seqs)]
    for res in fold.map(seqs):
        print(res)
    for f in structures.path.glob("*.xyz"):
        f.unlink()
