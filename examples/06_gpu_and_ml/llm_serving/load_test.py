# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/load_test.py", "--users", "8", "--requests-per-user", "3"]
# ---
# # Load testing the OpenAI-compatible server
#
# The locust analog: N concurrent user sessions (client threads, like locust
# users) hammer ONE serving pool — `@modal.concurrent(max_inputs=32)` lets a
# single replica interleave them, continuous batching does the rest.
# Latency percentiles land in a CSV on a Volume.

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


def user_session(target, user_id: int, n_requests: int) -> list:
    lat = []
    for i in range(n_requests):
        t0 = time.perf_counter()
        target.chat.remote(f"user {user_id} message {i} about wavefronts", 16)
        lat.append(time.perf_counter() - t0)
    return lat


@app.local_entrypoint()
def main(users: int = 8, requests_per_user: int = 3):
    import concurrent.futures

    target = Target()
    target.chat.remote("warmup", 4)  # model load outside the timed window
    t0 = time.perf_counter()
    with concurrent.futures.ThreadPoolExecutor(users) as pool:
        futs = [pool.submit(user_session, target, u, requests_per_user)
                for u in range(users)]
        all_lat = sorted(x for f in futs for x in f.result())
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
