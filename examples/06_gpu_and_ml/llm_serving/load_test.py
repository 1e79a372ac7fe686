# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/load_test.py", "--users", "8", "--requests-per-user", "3"]
# ---
# # Load testing the OpenAI-compatible server
#
# The locust-on-Modal analog: N concurrent client workers hammer the chat
# endpoint, latency percentiles land in a CSV on a Volume.

import time

import modal_examples_amd as modal

app = modal.App("example-load-test")

results_vol = modal.Volume.from_name("load-test-results", create_if_missing=True)


@app.cls(gpu="mi355x")
@modal.concurrent(max_inputs=32)
class Target:
    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        gpu = torch.cuda.is_available()
        cfg = LlamaConfig.llama3_8b() if gpu else LlamaConfig.small()
        eng = LlamaEngine(cfg, device="cuda" if gpu else "cpu",
                          dtype=torch.bfloat16 if gpu else torch.float32,
                          use_graph=gpu, kv_blocks=None if gpu else 256)
        self.server = LLMServer(eng)

    @modal.method()
    def chat(self, prompt: str, max_tokens: int = 16) -> str:
        return self.server.generate(prompt, max_tokens=max_tokens)


@app.function()
def user_session(user_id: int, n_requests: int) -> list:
    t = Target()
    lat = []
    for i in range(n_requests):
        t0 = time.perf_counter()
        t.chat.remote(f"user {user_id} message {i} about wavefronts", 16)
        lat.append(time.perf_counter() - t0)
    return lat


@app.local_entrypoint()
def main(users: int = 8, requests_per_user: int = 3):
    t0 = time.perf_counter()
    all_lat = sorted(
        x for lats in user_session.map(range(users), [requests_per_user] * users)
        for x in lats)
    wall = time.perf_counter() - t0
    n = len(all_lat)
    stats = {
        "requests": n,
        "rps": round(n / wall, 2),
        "p50_s": round(all_lat[n // 2], 3),
        "p90_s": round(all_lat[min(n - 1, int(n * 0.9))], 3),
        "p99_s": round(all_lat[min(n - 1, int(n * 0.99))], 3),
    }
    print(stats)
    csv = "metric,value\n" + "\n".join(f"{k},{v}" for k, v in stats.items())
    (results_vol.path / "latest.csv").write_text(csv)
    results_vol.commit()
    print("wrote", results_vol.path / "latest.csv")
