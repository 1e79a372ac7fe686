# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/engine_build.py"]
# ---
# # Engine build → fast boot (the TRT-LLM engine workflow role)
#
# The reference's trtllm pair separates a BUILD step (compile an engine with
# explicit max_batch/kv settings, store the artifact) from serving (boot the
# prebuilt engine in <30 s, trtllm_throughput.py:250).  MI355X mapping: the
# "engine" is the hipGraph-captured decode step + the paged-KV configuration;
# the build step validates the config on hardware, captures the graph once,
# and writes the build manifest + weights to a Volume; serving boots from the
# artifact and reports boot time.

import json
import time

import modal_examples_amd as modal

app = modal.App("example-engine-build")

engines = modal.Volume.from_name("llm-engines", create_if_missing=True)


def _mk_engine(torch, cfg_name, max_batch, kv_blocks):
    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig

    gpu = torch.cuda.is_available()
    cfg = LlamaConfig.llama3_8b() if gpu and cfg_name == "llama3-8b" else LlamaConfig.small()
    return LlamaEngine(cfg, device="cuda" if gpu else "cpu",
                       dtype=torch.bfloat16 if gpu else torch.float32,
                       max_batch=max_batch, use_graph=gpu,
                       kv_blocks=kv_blocks if not gpu else None)


@app.function(gpu="mi355x", timeout=1800)
def build_engine(name: str = "llama3-8b", max_batch: int = 64,
                 kv_blocks: int = 128) -> dict:
    """The engine-build step: validate + capture + persist the artifact."""
    import torch

    t0 = time.time()
    eng = _mk_engine(torch, name, max_batch, kv_blocks)
    # warm the captured decode path once (the "compile")
    rid = eng.add_request([1, 2, 3], max_new_tokens=4)
    while rid not in eng.finished:
        eng.step()
    build_s = time.time() - t0
    art_dir = engines.path / name
    art_dir.mkdir(parents=True, exist_ok=True)
    torch.save(eng.model.state_dict(), art_dir / "weights.pt")
    manifest = {
        "name": name, "max_batch": max_batch,
        "kv_blocks": int(eng.num_blocks), "build_s": round(build_s, 2),
        "hipgraph": bool(eng.use_graph),
        "params_b": round(sum(p.numel() for p in eng.model.parameters()) / 1e9, 3),
    }
    (art_dir / "manifest.json").write_text(json.dumps(manifest))
    eng.close()
    engines.commit()
    return manifest


@app.cls(gpu="mi355x", timeout=600)
class EngineServer:
    name: str = modal.parameter(default="llama3-8b")

    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.server import LLMServer

        t0 = time.time()
        engines.reload()
        man = json.loads((engines.path / self.name / "manifest.json").read_text())
        eng = _mk_engine(torch, man["name"], man["max_batch"], man["kv_blocks"])
        sd = torch.load(engines.path / self.name / "weights.pt",
                        map_location=str(eng.device))
        eng.model.load_state_dict(sd)
        self.server = LLMServer(eng, model_name=self.name)
        self.boot_s = time.time() - t0
        self.manifest = man

    @modal.method()
    def info(self) -> dict:
        return {"boot_s": round(self.boot_s, 2), **self.manifest}

    @modal.method()
    def chat(self, prompt: str, max_tokens: int = 16) -> str:
        return self.server.generate(prompt, max_tokens=max_tokens)


@app.local_entrypoint()
def main():
    man = build_engine.remote()
    print("built:", man)
    srv = EngineServer(name=man["name"])
    out = srv.chat.remote("hello engine")
    info = srv.info.remote()
    print(f"boot {info['boot_s']}s (build was {man['build_s']}s); "
          f"chat -> {out[:48]!r}")
    assert info["boot_s"] < 60, info
    assert isinstance(out, str) and out
