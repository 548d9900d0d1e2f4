"""GPU kernel numerics: HIP kernels vs the numpy reference solver.

Every test compares the CDNA4 kernel path against the plain fp32 numpy
reference (models/_solver_ref.py) on the SAME bf16-rounded inputs, so
tolerances reflect bf16-storage rounding only (both sides accumulate fp32).
"""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs a GPU", allow_module_level=True)

from spark_sklearn_amd.models._solver_ref import (  # noqa: E402
    LOSS_HINGE, LOSS_LOGISTIC, LOSS_SQUARED, BatchPlan, auto_lr,
    linear_fit_ref, linear_scores_ref)
from spark_sklearn_amd.ops import load_extension  # noqa: E402
from spark_sklearn_amd.ops.device_executor import (  # noqa: E402
    DeviceLinearExecutor, ensure_layout)


def _bf16_round(X):
    return torch.as_tensor(X).to(torch.bfloat16).float().numpy()


def _mkplan(B, loss, lr, l2=None, l1=None, max_iter=20, tol=0.0, mom=0.9,
            target=None, col_to_cand=None, n_classes=1):
    mk = lambda v, dt: np.full(B, v, dtype=dt)  # noqa: E731
    plan = BatchPlan(
        loss=np.asarray(loss, np.int32) * np.ones(B, np.int32),
        lr=np.asarray(lr, np.float32) * np.ones(B, np.float32),
        l1=np.zeros(B, np.float32) if l1 is None else np.asarray(l1, np.float32),
        l2=np.zeros(B, np.float32) if l2 is None else np.asarray(l2, np.float32),
        momentum=mk(mom, np.float32), tol=mk(tol, np.float32),
        max_iter=mk(max_iter, np.int32),
        target_class=(np.asarray(target, np.int32) if target is not None
                      else mk(1, np.int32)),
        col_to_cand=(np.asarray(col_to_cand, np.int32)
                     if col_to_cand is not None
                     else np.arange(B, dtype=np.int32)),
        n_classes=n_classes, fit_intercept=mk(1, np.uint8))
    return plan


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda", 0)


@pytest.fixture(scope="module")
def problem():
    rng = np.random.default_rng(7)
    n, f = 8192, 96          # F not a multiple of 64: exercises padding
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
    y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)
    return X, y


def test_layout_probe(dev):
    layout = ensure_layout(dev)
    assert layout in (0, 1, 2, 3)
    ext = load_extension()
    assert ext.get_layout() == layout
    print(f"\nMFMA 16x16x32 bf16 operand k-map layout = {layout} "
          f"(AMAP={layout >> 1}, BMAP={layout & 1})")


def test_msq_matches(dev, problem):
    X, y = problem
    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    Xb = _bf16_round(X)
    ref = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    assert abs(ex.msq - ref) / ref < 1e-3


def test_linear_fit_matches_ref_logistic(dev, problem):
    X, y = problem
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    B = 5
    plan = _mkplan(B, LOSS_LOGISTIC, lr,
                   l2=np.array([0.0, 1e-4, 1e-3, 1e-2, 0.1], np.float32),
                   max_iter=12)
    train = np.arange(0, 6000, dtype=np.int64)

    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    (W_d, b_d), ni_d, loss_d = ex.fit(train, plan)
    W_r, b_r, ni_r, loss_r = linear_fit_ref(Xb[train], y[train], plan)

    assert W_d.shape == W_r.shape == (X.shape[1], B)
    np.testing.assert_allclose(W_d, W_r, rtol=0.05, atol=5e-4)
    np.testing.assert_allclose(b_d, b_r, rtol=0.05, atol=5e-4)
    np.testing.assert_allclose(loss_d, loss_r, rtol=0.02)
    assert list(ni_d) == list(ni_r)


def test_linear_fit_sample_weight_matches_ref(dev, problem):
    """Weighted fit (WT=1 fwd epilogue) vs the weighted numpy reference;
    also: unit weights reproduce the unweighted fit exactly."""
    X, y = problem
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    rng = np.random.default_rng(11)
    w_full = rng.uniform(0.2, 3.0, size=X.shape[0]).astype(np.float64)
    w_full = (w_full * (len(w_full) / w_full.sum())).astype(np.float32)
    lr = auto_lr(msq * float(w_full.max()), LOSS_LOGISTIC, 0.0)
    B = 4
    plan = _mkplan(B, LOSS_LOGISTIC, lr,
                   l2=np.array([0.0, 1e-4, 1e-3, 1e-2], np.float32),
                   max_iter=12)
    train = np.arange(0, 6000, dtype=np.int64)

    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    (W_d, b_d), ni_d, loss_d = ex.fit(train, plan, sample_weight=w_full)
    W_r, b_r, ni_r, loss_r = linear_fit_ref(
        Xb[train], y[train], plan, sample_weight=w_full[train])
    np.testing.assert_allclose(W_d, W_r, rtol=0.05, atol=5e-4)
    np.testing.assert_allclose(b_d, b_r, rtol=0.05, atol=5e-4)
    np.testing.assert_allclose(loss_d, loss_r, rtol=0.02)
    assert list(ni_d) == list(ni_r)

    # unit weights == unweighted, bit-for-bit on device outputs
    ones = np.ones(X.shape[0], np.float32)
    (W_1, b_1), _, _ = ex.fit(train, plan, sample_weight=ones)
    (W_0, b_0), _, _ = ex.fit(train, plan)
    np.testing.assert_allclose(W_1, W_0, rtol=2e-3, atol=1e-5)
    np.testing.assert_allclose(b_1, b_0, rtol=2e-3, atol=1e-5)


def test_linear_fit_l1_and_hinge(dev, problem):
    X, y = problem
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    B = 4
    plan = _mkplan(B, [LOSS_LOGISTIC, LOSS_LOGISTIC, LOSS_HINGE, LOSS_HINGE],
                   [auto_lr(msq, LOSS_LOGISTIC, 0.0)] * 2 +
                   [auto_lr(msq, LOSS_HINGE, 0.0) * 0.1] * 2,
                   l1=np.array([0.0, 1e-3, 0.0, 0.0], np.float32),
                   l2=np.array([0.0, 0.0, 1e-3, 1e-2], np.float32),
                   max_iter=10)
    train = np.arange(100, 7000, dtype=np.int64)

    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    (W_d, b_d), _, _ = ex.fit(train, plan)
    W_r, b_r, _, _ = linear_fit_ref(Xb[train], y[train], plan)
    np.testing.assert_allclose(W_d, W_r, rtol=0.08, atol=1e-3)
    np.testing.assert_allclose(b_d, b_r, rtol=0.08, atol=1e-3)


def test_early_stop_n_iter(dev, problem):
    X, y = problem
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    plan = _mkplan(2, LOSS_LOGISTIC, lr, max_iter=200)
    plan.tol[:] = [2e-3, 0.0]
    train = np.arange(4096, dtype=np.int64)
    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    _, ni_d, _ = ex.fit(train, plan)
    assert ni_d[0] < ni_d[1]
    assert ni_d[1] == 200


def test_scores_match_ref(dev, problem):
    X, y = problem
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    B = 6
    plan = _mkplan(B, LOSS_LOGISTIC, lr,
                   l2=np.logspace(-5, -1, B).astype(np.float32),
                   max_iter=15)
    train = np.arange(0, 6000, dtype=np.int64)
    test = np.arange(6000, 8192, dtype=np.int64)

    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    (W_d, b_d), _, _ = ex.fit(train, plan)
    for metric in ("accuracy", "neg_log_loss"):
        s_d = ex.scores(test, (W_d, b_d), plan, metric)
        s_r = linear_scores_ref(Xb[test], y[test],
                                W_d, b_d, plan, metric)
        np.testing.assert_allclose(s_d, s_r, rtol=5e-3, atol=5e-3)


def test_r2_scores(dev):
    rng = np.random.default_rng(11)
    n, f = 4096, 64
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f).astype(np.float32)
    yr = X @ w + 0.1 * rng.standard_normal(n).astype(np.float32)
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_SQUARED, 0.0)
    plan = _mkplan(2, LOSS_SQUARED, lr, max_iter=60, target=[-1, -1])
    train = np.arange(0, 3000, dtype=np.int64)
    test = np.arange(3000, 4096, dtype=np.int64)
    ex = DeviceLinearExecutor(X, yr, None, dev)
    (W_d, b_d), _, _ = ex.fit(train, plan)
    s_d = ex.scores(test, (W_d, b_d), plan, "r2")
    s_r = linear_scores_ref(Xb[test], yr[test], W_d, b_d, plan, "r2")
    np.testing.assert_allclose(s_d, s_r, rtol=1e-3, atol=1e-3)
    assert s_d[0] > 0.9


def test_multiclass_ovr_device(dev):
    rng = np.random.default_rng(5)
    n, f, C = 6000, 64, 4
    centers = rng.standard_normal((C, f)).astype(np.float32) * 2
    y = rng.integers(0, C, n)
    X = (centers[y] + rng.standard_normal((n, f))).astype(np.float32)
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    # 2 candidates x 4 classes = 8 columns
    plan = _mkplan(8, LOSS_LOGISTIC, lr,
                   l2=np.repeat([1e-5, 1e-2], 4).astype(np.float32),
                   max_iter=20, target=list(range(4)) * 2,
                   col_to_cand=[0] * 4 + [1] * 4, n_classes=4)
    train = np.arange(0, 4500, dtype=np.int64)
    test = np.arange(4500, 6000, dtype=np.int64)
    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    (W_d, b_d), _, _ = ex.fit(train, plan)
    W_r, b_r, _, _ = linear_fit_ref(Xb[train], y[train], plan)
    np.testing.assert_allclose(W_d, W_r, rtol=0.08, atol=1e-3)
    acc_d = ex.scores(test, (W_d, b_d), plan, "accuracy")
    acc_r = linear_scores_ref(Xb[test], y[test], W_d, b_d, plan, "accuracy")
    np.testing.assert_allclose(acc_d, acc_r, atol=5e-3)
    assert acc_d[0] > 0.9


def test_full_grid_search_on_device(dev, problem):
    """End-to-end GridSearchCV with the engine on GPU vs CPU tier."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    X, y = problem
    grid = {"C": [0.01, 0.1, 1.0, 10.0]}
    gs_gpu = ska.GridSearchCV(None, LogisticRegression(max_iter=15, tol=0.0),
                              grid, cv=3).fit(X, y)
    # CPU tier on the same machine: force host execution via device-less run
    import spark_sklearn_amd.engine as eng

    classes, y_enc = eng.LinearFamily.encode_targets(
        LogisticRegression(), y)
    host = eng._HostLinearExecutor(_bf16_round(X), y_enc, classes)
    assert gs_gpu.best_score_ > 0.85
    # scores across the grid agree with the host tier to bf16 tolerance
    from sklearn.model_selection import KFold

    cv = list(KFold(3).split(X))
    from spark_sklearn_amd.engine import LinearFamily

    plan, _ = LinearFamily.build_plan(LogisticRegression(max_iter=15, tol=0.0),
                                      gs_gpu.cv_results_["params"],
                                      len(cv[0][0]), host, 2, "raise")
    fitted, _, _ = host.fit(cv[0][0], plan)
    s_host = host.scores(cv[0][1], fitted, plan, "accuracy")
    np.testing.assert_allclose(
        gs_gpu.cv_results_["split0_test_score"], s_host, atol=0.02)


def test_extension_mandatory_on_gpu(dev):
    """The ops loader must not silently fall back on a GPU box."""
    from spark_sklearn_amd.ops import extension_available

    assert extension_available(), \
        "native extension missing on a GPU box -- hard failure by policy"


@pytest.mark.parametrize("f", [96, 256])
def test_wide_batch_bp256_matches_ref(dev, f):
    """Bp=256 kernel paths (fwd_d7 + bwd_g5/g8) vs the numpy reference.

    Regression guard: the wide-batch kernels the flagship bench runs were
    originally only numerics-tested at Bp=64 (different kernels); a
    staging-coverage bug in bwd_g5 shipped undetected until a full-width
    test existed."""
    rng = np.random.default_rng(3)
    n = 4096
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
    y = ((X @ w) > 0).astype(np.int64)
    Xb = _bf16_round(X)
    msq = float(np.mean(np.sum(Xb.astype(np.float64) ** 2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    B = 256
    plan = _mkplan(B, LOSS_LOGISTIC, lr,
                   l2=np.logspace(-6, -2, B).astype(np.float32),
                   max_iter=6)
    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, dev)
    train = np.arange(0, 3500, dtype=np.int64)
    (W_d, b_d), ni_d, _ = ex.fit(train, plan)
    W_r, b_r, ni_r, _ = linear_fit_ref(Xb[train], y[train], plan)
    assert np.isfinite(W_d).all()
    np.testing.assert_allclose(W_d, W_r, rtol=0.05, atol=2e-4)
    np.testing.assert_allclose(b_d, b_r, rtol=0.05, atol=2e-4)
    assert list(ni_d) == list(ni_r)
    s_d = ex.scores(train, (W_d, b_d), plan, "accuracy")
    s_r = linear_scores_ref(Xb[train], y[train], W_d, b_d, plan, "accuracy")
    np.testing.assert_allclose(s_d, s_r, atol=5e-3)


def test_native_search_run_matches_per_chunk(dev):
    """The C++ whole-search runner must equal the per-chunk path."""
    import spark_sklearn_amd.engine as eng
    from spark_sklearn_amd.base_search import Task
    from spark_sklearn_amd.models import LogisticRegression
    from sklearn.model_selection import KFold

    rng = np.random.default_rng(11)
    n, f = 8192, 96
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
    y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)
    est = LogisticRegression(max_iter=15, tol=0.0)
    splits = list(KFold(3).split(X))
    tasks = []
    i = 0
    for ci, params in enumerate([{"C": c} for c in (0.01, 0.1, 1.0, 10.0)]):
        for fi, (tr, te) in enumerate(splits):
            tasks.append(Task(i, ci, fi, params, np.asarray(tr),
                              np.asarray(te)))
            i += 1

    def run(disable_fast):
        orig = eng._linear_device_search
        if disable_fast:
            eng._linear_device_search = lambda *a, **k: None
        try:
            res = eng.run_batched_tasks(
                est, X, y, tasks, scorer=None, fit_params=None,
                return_train_score=True, error_score="raise", verbose=0)
        finally:
            eng._linear_device_search = orig
        return sorted(res, key=lambda r: r.index)

    slow, fast = run(True), run(False)
    for s_, f_ in zip(slow, fast):
        assert abs(s_.test_score - f_.test_score) < 2e-3, \
            (s_.index, s_.test_score, f_.test_score)
        assert abs(s_.train_score - f_.train_score) < 2e-3


def test_csr_resident_kernels_match_ref(dev):
    """K5 CSR-resident fit/scores vs the numpy reference on the same
    bf16-rounded sparse matrix (never densified on device)."""
    import scipy.sparse as sp

    from spark_sklearn_amd.ops.sparse_executor import SparseLinearExecutor

    rng = np.random.default_rng(5)
    n, f = 20000, 512
    mask = rng.random((n, f)) < 0.02
    Xd = (rng.standard_normal((n, f)) * mask).astype(np.float32)
    Xs = sp.csr_matrix(Xd)
    w = rng.standard_normal(f).astype(np.float32)
    y = ((Xd @ w) > 0).astype(np.int64)
    Xb = _bf16_round(Xd)

    sq = np.sum(Xb.astype(np.float64) ** 2, axis=1)
    msq = float(np.mean(sq))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    B = 96
    plan = _mkplan(B, LOSS_LOGISTIC, lr,
                   l2=np.logspace(-6, -2, B).astype(np.float32),
                   max_iter=8)
    ex = SparseLinearExecutor(Xs, y.astype(np.float32), None, dev)
    train = np.arange(0, 15000, dtype=np.int64)
    test = np.arange(15000, 20000, dtype=np.int64)
    (W_d, b_d), ni_d, _ = ex.fit(train, plan)
    W_r, b_r, ni_r, _ = linear_fit_ref(Xb[train], y[train], plan)
    assert np.isfinite(W_d).all()
    np.testing.assert_allclose(W_d, W_r, rtol=0.05, atol=2e-4)
    np.testing.assert_allclose(b_d, b_r, rtol=0.05, atol=2e-4)
    assert list(ni_d) == list(ni_r)
    s_d = ex.scores(test, (W_d, b_d), plan, "accuracy")
    s_r = linear_scores_ref(Xb[test], y[test], W_d, b_d, plan, "accuracy")
    np.testing.assert_allclose(s_d, s_r, atol=5e-3)


def test_csr_grid_search_end_to_end(dev, monkeypatch):
    """Low-density sparse input routes through the CSR kernels inside a
    full GridSearchCV (capacity gate forced open: the default now picks
    dense-ingest whenever the dense form fits -- see
    profiles/r02_csr_crossover.md)."""
    import scipy.sparse as sp

    import spark_sklearn_amd as ska
    import spark_sklearn_amd.ops.sparse_executor as se
    from spark_sklearn_amd.models import LogisticRegression

    monkeypatch.setattr(se, "CSR_MIN_DENSE_BYTES", 0)
    rng = np.random.default_rng(6)
    n, f = 30000, 256
    mask = rng.random((n, f)) < 0.02
    Xd = (rng.standard_normal((n, f)) * mask).astype(np.float32)
    Xs = sp.csr_matrix(Xd)
    w = rng.standard_normal(f).astype(np.float32)
    y = ((Xd @ w) > 0).astype(np.int64)
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=20, tol=0.0),
                          {"C": [0.1, 1.0, 10.0]}, cv=3,
                          return_train_score=False).fit(Xs, y)
    assert gs.best_score_ > 0.8


def test_keyed_batched_fits_match_per_key(dev):
    """KeyedEstimator GPU fast path (keys as masked columns) must match
    per-key individual fits."""
    import pandas as pd

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(9)
    rows = []
    for k in range(6):
        w = rng.standard_normal(16)
        n_k = int(rng.integers(300, 800))
        Xk = rng.standard_normal((n_k, 16))
        yk = (Xk @ w > 0).astype(np.int64)
        for x, yy in zip(Xk, yk):
            rows.append({"k": k, "features": x, "label": int(yy)})
    df = pd.DataFrame(rows)

    est = LogisticRegression(max_iter=30, tol=0.0)
    ke = ska.KeyedEstimator(sklearnEstimator=est, keyCols=["k"],
                            xCol="features", yCol="label")
    batched = ke._fit_keys_batched(
        [(k if isinstance(k, tuple) else (k,), g)
         for k, g in df.groupby(["k"], sort=True)])
    assert batched is not None, "fast path did not engage"
    model = ke.fit(df)
    out = model.transform(df)

    # oracle: per-key reference fits with IDENTICAL hyperparameters --
    # the batched path derives auto-lr from the stacked matrix's mean
    # squared row norm, so the oracle must use the same value (comparing
    # against per-key auto-lr would conflate lr choice with correctness)
    X_all = np.stack(df["features"].values).astype(np.float32)
    Xb_all = _bf16_round(X_all)
    msq_global = float(np.mean(np.sum(Xb_all.astype(np.float64) ** 2,
                                      axis=1)))
    for k, group in df.groupby("k"):
        X = np.stack(group["features"].values).astype(np.float32)
        y = group["label"].values.astype(np.int64)
        Xb = _bf16_round(X)
        spec = LogisticRegression(max_iter=30, tol=0.0)._column_spec(
            len(X), msq_global)
        plan = _mkplan(1, spec["loss"], spec["lr"],
                       l2=np.array([spec["l2"]], np.float32),
                       max_iter=30)
        W_r, b_r, _, _ = linear_fit_ref(Xb, y, plan)
        batched_est = None
        for rec in batched:
            if rec[0] == k:
                batched_est = rec[-1].estimator
        assert batched_est is not None
        np.testing.assert_allclose(batched_est.coef_.ravel(), W_r[:, 0],
                                   rtol=0.05, atol=3e-4)
        got = out.loc[group.index, "prediction"].values
        pred_ref = (Xb @ W_r[:, 0] + b_r[0] > 0).astype(int)
        agree = np.mean(got == pred_ref)
        assert agree > 0.99, (k, agree)


def test_multimetric_device_search(problem):
    """Multimetric scoring on the GPU: every metric device-computable ->
    per-metric executor.scores; results match two single-metric runs."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    X, y = problem
    est = LogisticRegression(max_iter=20, tol=0.0)
    gm = ska.GridSearchCV(None, est, {"C": [0.1, 1.0, 10.0]},
                          scoring=["accuracy", "neg_log_loss"],
                          refit="accuracy", cv=3).fit(X, y)
    assert "mean_test_neg_log_loss" in gm.cv_results_
    g1 = ska.GridSearchCV(None, est, {"C": [0.1, 1.0, 10.0]},
                          scoring="accuracy", cv=3).fit(X, y)
    np.testing.assert_allclose(gm.cv_results_["mean_test_accuracy"],
                               g1.cv_results_["mean_test_score"],
                               rtol=1e-6)
    g2 = ska.GridSearchCV(None, est, {"C": [0.1, 1.0, 10.0]},
                          scoring="neg_log_loss", cv=3).fit(X, y)
    # per-chunk path vs native fast path: fp32 sum-order differs
    np.testing.assert_allclose(gm.cv_results_["mean_test_neg_log_loss"],
                               g2.cv_results_["mean_test_score"],
                               rtol=1e-6)
    assert gm.best_score_ == g1.best_score_


def test_determinism_and_stability(problem):
    """Race canary: two identical searches on the
    same resident data must agree to float-atomic reordering noise
    (split-K G-sums make bit-exactness impossible by design; scores must
    match to ~1e-6 and the ranking exactly)."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    X, y = problem
    est = LogisticRegression(max_iter=15, tol=0.0)
    grid = {"C": [0.05, 0.5, 5.0]}
    a = ska.GridSearchCV(None, est, grid, cv=3,
                         return_train_score=False).fit(X, y)
    b = ska.GridSearchCV(None, est, grid, cv=3,
                         return_train_score=False).fit(X, y)
    np.testing.assert_allclose(a.cv_results_["mean_test_score"],
                               b.cv_results_["mean_test_score"],
                               rtol=0, atol=2e-5)
    assert list(a.cv_results_["rank_test_score"]) == \
        list(b.cv_results_["rank_test_score"])
    assert a.best_params_ == b.best_params_


def test_gbt_run_to_run_stability(problem):
    """Race canary for GBT: the g-histogram uses float LDS atomics, so
    tree structures may flip at ulp-tie splits across runs (measured --
    (h, count) are already deterministic fixed-point; LightGBM-GPU has
    the same property).  Model QUALITY must be stable: two fits on the
    same data agree on held-out accuracy to ~1e-3."""
    from spark_sklearn_amd.models._gbt_ref import GBTSpec
    from spark_sklearn_amd.models.gbt import _GBTPlan
    from spark_sklearn_amd.ops.gbt_executor import DeviceGBTExecutor

    X, y = problem
    dev = torch.device("cuda", 0)
    spec = GBTSpec(n_estimators=6, max_depth=4, learning_rate=0.3,
                   min_samples_leaf=20)
    train = np.arange(6000, dtype=np.int64)
    test = np.arange(6000, 8000, dtype=np.int64)
    accs = []
    for _ in range(2):
        ex = DeviceGBTExecutor(np.ascontiguousarray(X),
                               y.astype(np.float32), dev)
        plan = _GBTPlan([spec])
        fitted, _, _ = ex.fit(train, plan)
        accs.append(ex.scores(test, fitted, plan, "accuracy")[0])
    # accuracy granularity on 2000 held-out rows is 5e-4: allow a few
    # ulp-tie-flipped predictions (a REAL race yields garbage, not 0.3%)
    assert abs(accs[0] - accs[1]) < 5e-3


def test_cuda_tensor_input_matches_numpy(problem):
    """bench.py hands ranks a CUDA-resident X after the RC1 broadcast;
    that tensor input path must score identically to numpy input."""
    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task
    from spark_sklearn_amd.engine import run_batched_tasks
    from spark_sklearn_amd.models import LogisticRegression

    X, y = problem
    dev = torch.device("cuda", 0)
    Xt = torch.as_tensor(X).to(dev)
    est = LogisticRegression(max_iter=10, tol=0.0)
    splits = list(KFold(3).split(np.arange(len(X))))

    def run(data):
        tasks = []
        i = 0
        for ci, p in enumerate([{"C": 0.5}, {"C": 5.0}]):
            for fi, (tr, te) in enumerate(splits):
                tasks.append(Task(i, ci, fi, p, tr, te))
                i += 1
        out = run_batched_tasks(
            est, data, y, tasks, scorer=None, fit_params=None,
            return_train_score=False, error_score="raise", verbose=0,
            device=dev)
        return {r.index: r.test_score for r in out}

    a = run(X)
    b = run(Xt)
    assert a.keys() == b.keys()
    for k in a:
        assert abs(a[k] - b[k]) < 1e-12, (k, a[k], b[k])


def test_csr_multiclass_ovr_matches_dense(dev, monkeypatch):
    """CSR-resident OVR multiclass (round-2): same scores as the dense
    executor on the same data (CSR gate forced open)."""
    import scipy.sparse as sp
    import spark_sklearn_amd.ops.sparse_executor as se
    from spark_sklearn_amd.models import LogisticRegression

    monkeypatch.setattr(se, "CSR_MIN_DENSE_BYTES", 0)
    rng = np.random.default_rng(0)
    n, f, K = 40_000, 256, 4
    X = sp.random(n, f, density=0.02, format="csr", random_state=1,
                  dtype=np.float32)
    w = rng.standard_normal((f, K)).astype(np.float32)
    y = np.asarray((X @ w) + 0.1 * rng.standard_normal((n, K)),
                   dtype=np.float32).argmax(axis=1)

    grid = {"C": [0.1, 1.0, 10.0]}
    import spark_sklearn_amd as ska

    gs_sp = ska.GridSearchCV(None, LogisticRegression(max_iter=25, tol=0.0),
                             grid, cv=2)
    gs_sp.fit(X, y)
    gs_d = ska.GridSearchCV(None, LogisticRegression(max_iter=25, tol=0.0),
                            grid, cv=2)
    gs_d.fit(np.asarray(X.todense()), y)
    np.testing.assert_allclose(gs_sp.cv_results_["mean_test_score"],
                               gs_d.cv_results_["mean_test_score"],
                               atol=5e-3)
    assert gs_sp.best_score_ > 0.6


def test_multiclass_wide_tile_scoring_matches_host(dev):
    """scores8 (wide multiclass tile, round-2) vs the CPU engine on a
    test fold large enough to trigger the wide path (>= 4096 rows)."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(3)
    n, f, K = 30_000, 128, 5
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal((f, K)).astype(np.float32)
    y = (X @ w + 0.5 * rng.standard_normal((n, K))).argmax(axis=1)

    grid = {"C": [0.1, 1.0, 10.0]}
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=25, tol=0.0),
                          grid, cv=3)
    gs.fit(X, y)
    assert gs.cv_results_["mean_test_score"].shape == (3,)
    assert gs.best_score_ > 0.75
    gs_host = ska.GridSearchCV("sequential",
                               LogisticRegression(max_iter=25, tol=0.0),
                               grid, cv=3)
    gs_host.fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               gs_host.cv_results_["mean_test_score"],
                               atol=7e-3)


def test_class_weight_maps_to_sample_weight_on_device(dev, problem):
    """class_weight (engine-composed onto the weighted kernels) equals
    the manual sample_weight fit on the device tier."""
    from spark_sklearn_amd.engine import fit_single
    from spark_sklearn_amd.models import LogisticRegression as SkaLR

    X, y = problem
    cw = {0: 1.0, 1: 3.0}
    m1 = SkaLR(max_iter=8, class_weight=cw)
    fit_single(m1, X, y, device=dev)
    w = np.where(y == 1, 3.0, 1.0).astype(np.float64)
    m2 = SkaLR(max_iter=8)
    fit_single(m2, X, y, device=dev, sample_weight=w)
    np.testing.assert_allclose(m1.coef_, m2.coef_, rtol=1e-4, atol=1e-6)
