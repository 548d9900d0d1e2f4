import sys
sys.path.insert(0, '/root/repo')
import numpy as np
import spark_sklearn_amd.engine as eng
from spark_sklearn_amd.base_search import Task
from spark_sklearn_amd.models import LogisticRegression
from sklearn.model_selection import KFold

rng = np.random.default_rng(7)
n, f = 8192, 96
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)

est = LogisticRegression(max_iter=15, tol=0.0)
splits = list(KFold(3).split(X))
tasks = []
i = 0
for ci, params in enumerate([{"C": 0.01}, {"C": 0.1}, {"C": 1.0}, {"C": 10.0}]):
    for fi, (tr, te) in enumerate(splits):
        tasks.append(Task(i, ci, fi, params, np.asarray(tr), np.asarray(te)))
        i += 1

def run(disable_fast):
    orig = eng._linear_device_search
    if disable_fast:
        eng._linear_device_search = lambda *a, **k: None
    try:
        res = eng.run_batched_tasks(est, X, y, tasks, scorer=None,
                                    fit_params=None, return_train_score=True,
                                    error_score="raise", verbose=0)
    finally:
        eng._linear_device_search = orig
    return sorted(res, key=lambda r: r.index)

slow = run(True)
fast = run(False)
for s_, f_ in zip(slow, fast):
    print(f"task {s_.index}: slow {s_.test_score:.4f}/{s_.train_score:.4f}  fast {f_.test_score:.4f}/{f_.train_score:.4f}")
