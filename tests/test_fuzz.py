"""Hypothesis shape/edge fuzzing.

Properties over the HOST tier, whose numerics define the kernels:

* column independence (the core batching invariant: folding B
  candidates into one launch must not change any candidate's scores);
* the family references stay finite and well-formed across shapes.
"""

import numpy as np
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

import spark_sklearn_amd as ska  # noqa: E402


@settings(max_examples=20, deadline=None)
@given(n=st.integers(24, 200), f=st.integers(1, 48),
       n_classes=st.integers(2, 4), seed=st.integers(0, 10_000),
       penalty=st.sampled_from(["l1", "l2"]))
def test_fuzz_batched_equals_sequential_linear(n, f, n_classes, seed,
                                               penalty):
    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task, fit_and_score_task
    from spark_sklearn_amd.engine import run_batched_tasks
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = rng.integers(0, n_classes, n)
    if len(np.unique(y)) < n_classes:       # degenerate draw: skip
        return
    est = LogisticRegression(max_iter=8, tol=0.0, penalty=penalty)
    splits = list(KFold(2).split(np.arange(n)))
    cands = [{"C": 0.5}, {"C": 2.0}]
    tasks = []
    i = 0
    for ci, p in enumerate(cands):
        for fi, (tr, te) in enumerate(splits):
            tasks.append(Task(i, ci, fi, p, tr, te))
            i += 1
    def run(task_list):
        out = run_batched_tasks(
            est, X, y, task_list, scorer=None, fit_params=None,
            return_train_score=False, error_score="raise", verbose=0,
            device=None)
        return {r.index: r.test_score for r in out}

    # column-independence invariant: candidates batched together score
    # EXACTLY as each candidate batched alone (the kernels fold B
    # candidates into one launch; no cross-column contamination allowed)
    together = run(tasks)
    for ci in range(len(cands)):
        alone = run([t for t in tasks if t.cand_index == ci])
        for idx, v in alone.items():
            assert together[idx] == v, (cands[ci], idx)

    # and the scores agree with a standalone per-task fit up to the
    # documented lr divergence (auto-lr uses the full-matrix msq in the
    # batched search, the train-subset msq in a standalone .fit())
    t0 = tasks[0]
    single = fit_and_score_task(est, X, y, _acc_scorer, t0,
                                return_train_score=False,
                                error_score="raise")
    assert 0.0 <= single.test_score <= 1.0
    assert 0.0 <= together[t0.index] <= 1.0


def _acc_scorer(est, X, y):
    return float(np.mean(est.predict(X) == y))


@settings(max_examples=20, deadline=None)
@given(n=st.integers(30, 300), f=st.integers(1, 24),
       depth=st.integers(1, 4), trees=st.integers(1, 8),
       seed=st.integers(0, 10_000))
def test_fuzz_gbt_reference_shapes(n, f, depth, trees, seed):
    from spark_sklearn_amd.models._gbt_ref import (
        GBTSpec, bin_data, gbt_fit_ref, gbt_predict_raw, quantile_bins)

    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    edges = quantile_bins(X)
    Xb = bin_data(X, edges)
    spec = GBTSpec(n_estimators=trees, max_depth=depth,
                   learning_rate=0.3, min_samples_leaf=1)
    fitted, f0 = gbt_fit_ref(Xb, y, spec)
    assert len(fitted) == trees
    raw = gbt_predict_raw(Xb, fitted, f0)
    assert raw.shape == (n,) and np.all(np.isfinite(raw))


@settings(max_examples=20, deadline=None)
@given(n=st.integers(20, 300), f=st.integers(1, 24),
       k=st.integers(1, 8), iters=st.integers(1, 10),
       seed=st.integers(0, 10_000))
def test_fuzz_kmeans_reference_shapes(n, f, k, iters, seed):
    from spark_sklearn_amd.models._kmeans_ref import (
        KMeansPlan, kmeans_fit_ref, kmeans_init_centers)

    if k > n:
        return
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    C0 = kmeans_init_centers(X, k, "k-means++", rng)
    plan = KMeansPlan(cstart=np.array([0, k], np.int32),
                      block_to_cand=np.array([0], np.int32),
                      max_iter=np.array([iters], np.int32),
                      tol_scaled=np.array([0.0], np.float32), n_cand=1)
    C, n_iter, inertia = kmeans_fit_ref(X, C0, plan)
    assert C.shape == (k, f)
    assert np.all(np.isfinite(C)) and np.isfinite(inertia[0])
    assert inertia[0] >= 0.0
    assert 1 <= n_iter[0] <= iters


@settings(max_examples=15, deadline=None)
@given(n=st.integers(40, 150), f=st.integers(2, 16),
       seed=st.integers(0, 10_000))
def test_fuzz_search_never_breaks_contract(n, f, seed):
    """Whole-search property: arbitrary small problems always produce a
    complete, sklearn-shaped cv_results_."""
    from sklearn.svm import LinearSVC

    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.2 * rng.standard_normal(n) > 0).astype(np.int64)
    gs = ska.GridSearchCV("sequential", LinearSVC(), {"C": [0.1, 1.0]},
                          cv=2, return_train_score=False).fit(X, y)
    res = gs.cv_results_
    assert len(res["params"]) == 2
    for key in ("mean_test_score", "std_test_score", "rank_test_score",
                "mean_fit_time", "param_C"):
        assert len(res[key]) == 2
    assert set(res["rank_test_score"]) <= {1, 2}
    assert 0.0 <= gs.best_score_ <= 1.0
