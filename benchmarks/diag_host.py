#!/usr/bin/env python3
"""Host-overhead breakdown of one flagship bench step.

Times the phases of engine._linear_device_search separately (plan build,
plan upload, native linear_search_run, result assembly) to locate the
wall-vs-device gap seen in bench.py (~91 ms wall vs ~82 ms device).
"""

import sys
import time

sys.path.insert(0, "/root/repo")

import numpy as np
import torch

from sklearn.model_selection import KFold

from spark_sklearn_amd.base_search import Task
from spark_sklearn_amd.engine import run_batched_tasks, _device_or_none
from spark_sklearn_amd.models import LogisticRegression


def main():
    rng = np.random.default_rng(0)
    n, f = 1_000_000, 256
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(f, dtype=np.float32) / np.sqrt(f)
    y = ((X @ w + 0.25 * rng.standard_normal(n, dtype=np.float32)) > 0
         ).astype(np.int64)
    splits = list(KFold(n_splits=5).split(np.arange(n)))
    est = LogisticRegression(max_iter=25, tol=0.0)
    device = _device_or_none(None)
    assert device is not None

    grid = [{"C": float(c), "penalty": p}
            for p in ("l2", "l1") for c in np.logspace(-3, 3, 128)]

    def make_tasks():
        tasks, t = [], 0
        for ci, params in enumerate(grid):
            for fi, (tr, te) in enumerate(splits):
                tasks.append(Task(t, ci, fi, params, tr, te))
                t += 1
        return tasks

    # warm caches / JIT
    for _ in range(2):
        run_batched_tasks(est, X, y, make_tasks(), scorer=None,
                          fit_params=None, return_train_score=False,
                          error_score="raise", verbose=0, device=device,
                          column_batch=256)
    torch.cuda.synchronize()

    # instrument: monkeypatch time around the native call
    import spark_sklearn_amd.engine as eng
    phases = {}
    orig_search = None

    from spark_sklearn_amd.ops import load_extension
    ext = load_extension()
    orig = ext.linear_search_run

    def timed_search(*a, **k):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = orig(*a, **k)
        torch.cuda.synchronize()
        phases["native_call"] = phases.get("native_call", 0.0) + \
            time.perf_counter() - t0
        return out

    ext.linear_search_run = timed_search
    try:
        for rep in range(3):
            t0 = time.perf_counter()
            tasks = make_tasks()
            phases["make_tasks"] = phases.get("make_tasks", 0.0) + \
                time.perf_counter() - t0
            t0 = time.perf_counter()
            run_batched_tasks(est, X, y, tasks, scorer=None,
                              fit_params=None, return_train_score=False,
                              error_score="raise", verbose=0,
                              device=device, column_batch=256)
            torch.cuda.synchronize()
            phases["run_batched"] = phases.get("run_batched", 0.0) + \
                time.perf_counter() - t0
    finally:
        ext.linear_search_run = orig

    reps = 3
    print("per-step (ms), 128-cand x5 folds:")
    for k, v in sorted(phases.items()):
        print(f"  {k:14s} {1000*v/reps:8.2f}")
    print(f"  python_side    "
          f"{1000*(phases['run_batched']-phases['native_call'])/reps:8.2f}")


if __name__ == "__main__":
    main()
