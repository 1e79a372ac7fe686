# ---
# cmd: ["python", "examples/02_building_containers/import_sklearn.py"]
# ---
# # Programmatic app.run() with a non-torch framework
#
# The import_sklearn.py role (reference: 02_building_containers/
# import_sklearn.py:51): an app driven by `with app.run():` from a plain
# `python file.py` invocation — no CLI — running a scikit-learn workload in a
# container whose image layers install the package.

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

import modal_examples_amd as modal  # noqa: E402

app = modal.App("example-import-sklearn")

image = modal.Image.debian_slim().pip_install("scikit-learn", "numpy")


@app.function(image=image)
def fit_and_score() -> float:
    import numpy as np
    from sklearn.ensemble import GradientBoostingClassifier
    from sklearn.model_selection import train_test_split

    rng = np.random.default_rng(0)
    X = rng.standard_normal((2000, 16))
    y = (X[:, 0] * 1.5 + X[:, 3] - X[:, 7] > 0).astype(int)
    Xt, Xv, yt, yv = train_test_split(X, y, random_state=0)
    clf = GradientBoostingClassifier(n_estimators=40, random_state=0)
    clf.fit(Xt, yt)
    return float(clf.score(Xv, yv))


if __name__ == "__main__":
    with modal.enable_output():
        with app.run():
            acc = fit_and_score.remote()
    assert acc > 0.9, acc
    print(f"held-out accuracy: {acc:.3f}")
