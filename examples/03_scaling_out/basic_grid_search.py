# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/03_scaling_out/basic_grid_search.py"]
# ---
# Parallel hyperparameter grid search with `.starmap` across the pool.

import modal_examples_amd as modal

app = modal.App("example-grid-search")


@app.function()
def fit(lr: float, depth: int) -> dict:
    # a stand-in objective with a known optimum at lr=0.1, depth=4
    score = 1.0 / (1 + abs(lr - 0.1) * 10 + abs(depth - 4))
    return {"lr": lr, "depth": depth, "score": round(score, 4)}


@app.local_entrypoint()
def main():
    grid = [(lr, d) for lr in (0.01, 0.1, 0.5) for d in (2, 4, 8)]
    results = list(fit.starmap(grid))
    best = max(results, key=lambda r: r["score"])
    print("best:", best)
    assert best["lr"] == 0.1 and best["depth"] == 4
