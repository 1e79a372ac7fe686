# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/binder_design/design_binders.py"]
# ---
# # Protein binder design (the binder-design package role)
#
# The reference ships binder design as an installable typed PACKAGE whose
# stages run as Modal functions (06_gpu_and_ml/binder-design).  Same shape
# here: `binder_design/` (py.typed, config/sequences/scoring modules) is
# added to the image; generations of candidates score in a `.map` fan-out;
# elites persist to a Volume; the run verifiably improves the population.

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent))

import modal_examples_amd as modal  # noqa: E402
from binder_design import DesignConfig, mutate, random_binder, score_binder  # noqa: E402

app = modal.App("example-binder-design")

results = modal.Volume.from_name("binder-designs", create_if_missing=True)


@app.function()
def score_candidate(binder: str, target: str) -> tuple:
    return binder, score_binder(binder, target)


@app.local_entrypoint()
def main():
    import json
    import random

    cfg = DesignConfig()
    rng = random.Random(cfg.seed)
    pop = [random_binder(cfg.binder_len, rng) for _ in range(cfg.population)]
    history = []
    for gen in range(cfg.generations):
        scored = sorted(
            score_candidate.starmap([(b, cfg.target) for b in pop], order_outputs=False),
            key=lambda t: -t[1])
        best = scored[0]
        history.append(round(best[1], 3))
        elites = [b for b, _ in scored[: cfg.population // 4]]
        pop = elites + [mutate(rng.choice(elites), cfg.mutation_rate, rng)
                        for _ in range(cfg.population - len(elites))]
        print(f"gen {gen}: best {best[1]:.3f} {best[0]}")
    (results.path / "best.json").write_text(json.dumps(
        {"binder": scored[0][0], "score": scored[0][1], "history": history}))
    results.commit()
    assert history[-1] >= history[0], history  # selection must not regress
    assert history[-1] > history[0] + 0.1, f"no improvement: {history}"
