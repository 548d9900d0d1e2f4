"""Deep-forest (depth 8/10) RF/ET benchmark at 1M rows.

Measures the device cap depth (10) at scale for BASELINE color:
GridSearchCV over RandomForestClassifier {max_depth 8,10} and
ExtraTreesClassifier {max_depth 10}, 50 trees, cv=3, 1M x 64.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import numpy as np  # noqa: E402

import spark_sklearn_amd as ska  # noqa: E402
from spark_sklearn_amd.models import (  # noqa: E402
    ExtraTreesClassifier, RandomForestClassifier)


def main():
    rng = np.random.default_rng(0)
    n, f = 1_000_000, 64
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(8)
    y = (X[:, :8] @ w + 0.3 * rng.standard_normal(n) > 0).astype(np.int64)

    for name, est, grid in [
        ("rf", RandomForestClassifier(n_estimators=50, random_state=0),
         {"max_depth": [8, 10], "max_features": [16]}),
        ("et", ExtraTreesClassifier(n_estimators=50, random_state=0),
         {"max_depth": [10], "max_features": [16]}),
    ]:
        gs = ska.GridSearchCV(None, est, grid, cv=3)
        t0 = time.perf_counter()
        gs.fit(X, y)
        dt = time.perf_counter() - t0
        n_cand = len(gs.cv_results_["params"])
        trees = sum(50 * 3 for _ in range(n_cand))
        print(f"{name}: {dt:.2f}s  {n_cand} cands x cv3 x 50 trees "
              f"= {trees} trees -> {trees / dt:.0f} trees/s  "
              f"best={gs.best_score_:.4f} params={gs.best_params_}")


if __name__ == "__main__":
    main()
