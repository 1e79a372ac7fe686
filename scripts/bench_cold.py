#!/usr/bin/env python3
"""Cold-start p50 at realistic weight residence (the second headline metric).

The r1 number timed first-call-after-random-init in ONE process; this bench
measures what the metric means: p50 over N FRESH processes each bringing up
the flagship model from host-resident weights (a saved state_dict in /dev/shm
— the warm-pool / snapshot-restore residence: weights in host DRAM, not
re-downloaded, not re-initialized).  Per run it reports time-to-first-output
(interpreter + torch import + weight load + H2D + graph capture + first
batch) and the weight-restore bandwidth.

Usage:  python scripts/bench_cold.py [--runs 5] [--model sdxl|llama]
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def prepare(model: str, shm: str) -> dict:
    import torch

    os.makedirs(shm, exist_ok=True)
    meta = {"model": model}
    from modal_examples_amd.gpu import fastload

    if model == "sdxl":
        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline

        pipe = SDXLPipeline(device="cpu", dtype=torch.bfloat16)
        fastload.save_file(dict(pipe.unet.state_dict()), f"{shm}/unet.safetensors")
        fastload.save_file(dict(pipe.vae.state_dict()), f"{shm}/vae.safetensors")
        meta["bytes"] = sum(os.path.getsize(f"{shm}/{f}")
                            for f in ("unet.safetensors", "vae.safetensors"))
    else:
        from modal_examples_amd.models.llama.model import LlamaConfig, LlamaModel

        m = LlamaModel(LlamaConfig.llama3_8b()).to(torch.bfloat16)
        fastload.save_file(dict(m.state_dict()), f"{shm}/llama.safetensors")
        meta["bytes"] = os.path.getsize(f"{shm}/llama.safetensors")
    # touch into page cache
    for f in os.listdir(shm):
        with open(os.path.join(shm, f), "rb") as fh:
            while fh.read(1 << 24):
                pass
    return meta


def child(model: str, shm: str) -> None:
    t_start = time.perf_counter()
    t_load = 0.0  # file->device portion of restore (llama path reports it)
    import torch

    t_import = time.perf_counter()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    if model == "sdxl":
        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline

        from modal_examples_amd.gpu import fastload

        t0 = time.perf_counter()
        # mmap + pinned-staged blob load (gpu/fastload.py); model built
        # EMPTY on device (meta init), weights assigned as blob views
        sd_u = fastload.load_file(f"{shm}/unet.safetensors", device=dev)
        sd_v = fastload.load_file(f"{shm}/vae.safetensors", device=dev)
        pipe = SDXLPipeline(device=dev, dtype=torch.bfloat16,
                            init_weights=False)
        pipe.unet.load_state_dict(sd_u, assign=True)
        pipe.vae.load_state_dict(sd_v, assign=True)
        if dev == "cuda":
            torch.cuda.synchronize()
        t_restore = time.perf_counter() - t0
        nbytes = sum(v.numel() * v.element_size() for v in sd_u.values())
        nbytes += sum(v.numel() * v.element_size() for v in sd_v.values())
        t1 = time.perf_counter()
        pipe.generate(["cold start probe"] * 4, steps=4)  # batch 4 = the headline config (and the shipped MIOpen find-db coverage)
        if dev == "cuda":
            torch.cuda.synchronize()
        t_first = time.perf_counter() - t1
    else:
        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        from modal_examples_amd.gpu import fastload

        import threading

        t0 = time.perf_counter()
        box = {}

        def build():  # engine alloc (hipMalloc of the KV pool dominates)
            t1 = time.perf_counter()
            box["eng"] = LlamaEngine(
                LlamaConfig.llama3_8b(), device=dev, dtype=torch.bfloat16,
                use_graph=(dev == "cuda"), init_weights=False)
            box["t_eng"] = time.perf_counter() - t1

        # overlap: KV-pool + empty-model allocation runs WHILE the weight
        # file streams through pinned staging on its own stream
        th = threading.Thread(target=build)
        th.start()
        sd = fastload.load_file(f"{shm}/llama.safetensors", device=dev)
        t_load = time.perf_counter() - t0
        th.join()
        eng, t_eng = box["eng"], box["t_eng"]
        t1 = time.perf_counter()
        eng.model.load_state_dict(sd, assign=True)
        if dev == "cuda":
            torch.cuda.synchronize()
        t_assign = time.perf_counter() - t1
        if dev == "cuda":
            torch.cuda.synchronize()
        t_restore = time.perf_counter() - t0
        nbytes = sum(v.numel() * v.element_size() for v in sd.values())
        srv = LLMServer(eng, model_name="cold-probe")
        t1 = time.perf_counter()
        srv.generate("cold start probe", max_tokens=4)
        srv.shutdown()
        t_first = time.perf_counter() - t1
    total = time.perf_counter() - t_start
    print(json.dumps({
        "cold_start_s": round(total, 3),
        "import_s": round(t_import - t_start, 3),
        "restore_s": round(t_restore, 3),
        "load_s": round(t_load, 3),
        "engine_init_s": round(locals().get("t_eng", 0.0), 3),
        "assign_s": round(locals().get("t_assign", 0.0), 3),
        "restore_gb_s": round(nbytes / t_restore / 1e9, 2),
        "first_call_s": round(t_first, 3),
        "weight_gb": round(nbytes / 1e9, 2)}), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--runs", type=int, default=5)
    ap.add_argument("--model", choices=("sdxl", "llama"), default="sdxl")
    ap.add_argument("--child", default=None, help=argparse.SUPPRESS)
    args = ap.parse_args()
    if args.child:
        child(args.model, args.child)
        return
    shm = os.environ.get("MODAL_AMD_COLD_DIR", f"/dev/shm/mxa_cold_{args.model}")
    meta = prepare(args.model, shm)
    samples = []
    for i in range(args.runs):
        r = subprocess.run(
            [sys.executable, os.path.abspath(__file__), "--model", args.model,
             "--child", shm], capture_output=True, text=True, timeout=600,
            cwd=REPO)
        line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
        if not line:
            print(f"run {i} failed:\n{r.stdout[-500:]}\n{r.stderr[-1500:]}")
            continue
        d = json.loads(line[-1])
        samples.append(d)
        print(f"run {i}: {d}")
    if not samples:
        sys.exit(1)
    colds = sorted(s["cold_start_s"] for s in samples)
    out = {
        "metric": f"{args.model} cold-start p50 (fresh process, host-resident weights)",
        "cold_start_p50_s": colds[len(colds) // 2],
        "cold_start_all_s": colds,
        "restore_gb_s_median": sorted(s["restore_gb_s"] for s in samples)[len(samples) // 2],
        "weight_gb": samples[0]["weight_gb"],
        "runs": len(samples),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
