# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/gbm_hyperopt.py"]
# ---
# # Gradient-boosting hyperparameter optimization
#
# The xgboost-optuna pattern: successive-halving style rounds of parallel
# trials, shared trial history in a `Dict`, best config promoted each round.

import modal_examples_amd as modal

app = modal.App("example-gbm-hyperopt")

trials = modal.Dict.from_name("gbm-trials", create_if_missing=True)


@app.function()
def run_trial(trial_id: int, lr: float, depth: int, n_est: int) -> dict:
    import numpy as np
    from sklearn.ensemble import GradientBoostingClassifier
    from sklearn.model_selection import train_test_split

    rng = np.random.default_rng(0)
    X = rng.standard_normal((1200, 12))
    y = ((X[:, 0] * X[:, 1] + 0.5 * X[:, 2] ** 2 + 0.1 * rng.standard_normal(1200)) > 0).astype(int)
    Xtr, Xte, ytr, yte = train_test_split(X, y, random_state=0)
    clf = GradientBoostingClassifier(learning_rate=lr, max_depth=depth,
                                     n_estimators=n_est, random_state=0)
    clf.fit(Xtr, ytr)
    score = float(clf.score(Xte, yte))
    result = {"trial": trial_id, "lr": lr, "depth": depth, "n_est": n_est,
              "score": score}
    trials[trial_id] = result
    return result


@app.local_entrypoint()
def main():
    trials.clear()
    grid = [(i, lr, d, n) for i, (lr, d, n) in enumerate(
        (lr, d, n) for lr in (0.05, 0.1, 0.3) for d in (2, 3) for n in (50, 100))]
    results = list(run_trial.starmap(grid))
    # halving round: rerun top-3 with more estimators
    top = sorted(results, key=lambda r: -r["score"])[:3]
    promoted = list(run_trial.starmap(
        [(100 + i, r["lr"], r["depth"], r["n_est"] * 2) for i, r in enumerate(top)]))
    best = max(promoted, key=lambda r: r["score"])
    print(f"{len(trials)} trials recorded; best: {best}")
    trials.clear()
