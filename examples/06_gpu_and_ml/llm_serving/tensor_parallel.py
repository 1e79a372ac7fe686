# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/tensor_parallel.py"]
# ---
# # Tensor-parallel big-model serving
#
# The big-model TP shape (reference: --tensor-parallel-size at
# llm-serving/vllm_inference.py:180, sglang --tp, TRT-LLM tensor_parallel_size):
# Megatron-style column→row sharding from `parallel/tp.py`, one all-reduce
# per transformer block over RCCL/xGMI.  On MI355X (288 GB HBM3E) TP only
# matters for the 70B-405B class — each rank below holds 1/world of the
# block's parameters and produces the exact full-model output.
#
# Runs world=2 on CPU over gloo here; the same code is RCCL on MI355X ranks.

import os
import subprocess
import sys

import modal_examples_amd as modal

app = modal.App("example-tensor-parallel")

RANK_MAIN = r"""
import json, os, torch, torch.distributed as dist
from modal_examples_amd.parallel.tp import TPGroup, shard_linear

rank, world = int(os.environ["RANK"]), int(os.environ["WORLD_SIZE"])
# RCCL needs one GPU per rank; fall back to gloo on a smaller box
backend = ("nccl" if torch.cuda.is_available()
           and torch.cuda.device_count() >= world else "gloo")
dist.init_process_group(backend, rank=rank, world_size=world)
device = f"cuda:{rank}" if backend == "nccl" else "cpu"
if backend == "nccl":
    torch.cuda.set_device(rank)

# one "70B-class" transformer MLP block, same full weights on every rank
torch.manual_seed(0)
d_model, d_ff, batch = 512, 2048, 4
up = torch.nn.Linear(d_model, d_ff).to(device)
down = torch.nn.Linear(d_ff, d_model).to(device)
x = torch.randn(batch, d_model, device=device)
torch.set_grad_enabled(False)
ref = down(torch.nn.functional.gelu(up(x), approximate="tanh"))

tp = TPGroup()
up_s = shard_linear(up, "column", tp)      # no comm in forward
down_s = shard_linear(down, "row", tp)     # ONE all-reduce
y = down_s(torch.nn.functional.gelu(up_s(x), approximate="tanh"))

full = sum(p.numel() for p in (*up.parameters(), *down.parameters()))
shard = sum(p.numel() for m in (up_s, down_s) for p in m.parameters())

# ---- act 2: the whole serving ENGINE under TP (engine `tp=` option):
# head-sharded blocks, per-rank KV-head cache shard, rank-identical
# logits -> the continuous-batching loop needs no extra broadcast
from modal_examples_amd.models.llama.engine import LlamaEngine
from modal_examples_amd.models.llama.model import (LlamaConfig, LlamaModel,
                                                   shard_llama_state)

cfg = LlamaConfig.small()
torch.manual_seed(0)
full_state = dict(LlamaModel(cfg).to(torch.bfloat16).state_dict())
eng = LlamaEngine(cfg, device=device, dtype=torch.bfloat16,
                  use_graph=(device != "cpu"), eos_id=-1, tp=tp)
eng.model.load_state_dict(shard_llama_state(full_state, cfg, rank, world))
eng.add_request(list(range(10, 22)), max_new_tokens=6, temperature=0.0)
eng.run_until_done(max_steps=100)
toks = eng.finished[1].out_tokens

out = {"rank": rank, "max_err": float((y - ref).abs().max()),
       "shard_frac": shard / full, "engine_tokens": toks}
print("TPRESULT " + json.dumps(out), flush=True)
dist.barrier()
dist.destroy_process_group()
"""


@app.local_entrypoint()
def main(world: int = 2):
    import json
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                    "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": str(port)})
        procs.append(subprocess.Popen([sys.executable, "-c", RANK_MAIN],
                                      env=env, stdout=subprocess.PIPE,
                                      text=True))
    results = {}
    for p in procs:
        out, _ = p.communicate(timeout=240)
        assert p.returncode == 0, f"rank failed:\n{out[-2000:]}"
        for line in out.splitlines():
            if line.startswith("TPRESULT "):
                r = json.loads(line[len("TPRESULT "):])
                results[r["rank"]] = r
    for rank in range(world):
        r = results[rank]
        print(f"rank {rank}: max |tp - full| = {r['max_err']:.2e}, "
              f"params held = {r['shard_frac']:.2%} of full")
        assert r["max_err"] < 1e-4
        assert abs(r["shard_frac"] - 1 / world) < 0.05
    # TP engine: every rank decoded the SAME tokens (no divergence)
    tok_sets = {tuple(results[rank]["engine_tokens"]) for rank in range(world)}
    assert len(tok_sets) == 1 and len(results[0]["engine_tokens"]) == 6
    print(f"TP={world}: exact full-model output from 1/{world} shards, "
          "one all-reduce per block; TP engine decoded "
          f"{results[0]['engine_tokens']} identically on every rank")
