"""GPU GBT kernels (K3) vs the numpy reference on identical binned data."""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs a GPU", allow_module_level=True)

from spark_sklearn_amd.models._gbt_ref import (  # noqa: E402
    GBTSpec, LOSS_SQUARED, bin_data, gbt_fit_ref, gbt_predict_raw,
    quantile_bins)
from spark_sklearn_amd.ops.gbt_executor import DeviceGBTExecutor  # noqa: E402
from spark_sklearn_amd.models.gbt import _GBTPlan  # noqa: E402


@pytest.fixture(scope="module")
def problem():
    rng = np.random.default_rng(0)
    n, f = 20000, 32
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = ((X[:, 0] * X[:, 1] + X[:, 2] +
          0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
    return X, y


def test_device_trees_match_reference(problem):
    """Same binned data, same spec -> identical tree structures and
    near-identical leaf values (fp32 sum-order only)."""
    X, y = problem
    dev = torch.device("cuda", 0)
    ex = DeviceGBTExecutor(X, y.astype(np.float32), dev)
    spec = GBTSpec(n_estimators=8, max_depth=4, learning_rate=0.3,
                   min_samples_leaf=20)
    plan = _GBTPlan([spec])
    train = np.arange(15000, dtype=np.int64)
    (models, edges), n_iter, _ = ex.fit(train, plan)
    trees_d, f0_d = models[0]

    Xb = bin_data(X, ex.edges)
    trees_r, f0_r = gbt_fit_ref(Xb[train], y[train].astype(np.float32),
                                spec)
    assert abs(f0_d - f0_r) < 1e-6
    agree = 0
    for (fd, td, vd), (fr, tr, vr) in zip(trees_d, trees_r):
        if np.array_equal(fd, fr) and np.array_equal(td, tr):
            agree += 1
            np.testing.assert_allclose(vd, vr, rtol=1e-3, atol=1e-5)
    # fp32 histogram sum-order can flip a near-tied split occasionally;
    # the vast majority of trees must be structurally identical
    assert agree >= 6, f"only {agree}/8 trees match the reference"


def test_device_weighted_trees_match_reference(problem):
    """sample_weight on device (grad-kernel w scaling) vs the weighted
    numpy reference: same structures, same weighted f0."""
    X, y = problem
    dev = torch.device("cuda", 0)
    rng = np.random.default_rng(4)
    w = rng.uniform(0.25, 4.0, size=X.shape[0]).astype(np.float64)
    w = (w * (len(w) / w.sum()))
    ex = DeviceGBTExecutor(X, y.astype(np.float32), dev)
    spec = GBTSpec(n_estimators=6, max_depth=4, learning_rate=0.3,
                   min_samples_leaf=20)
    plan = _GBTPlan([spec])
    train = np.arange(15000, dtype=np.int64)
    (models, edges), n_iter, _ = ex.fit(train, plan, sample_weight=w)
    trees_d, f0_d = models[0]

    Xb = bin_data(X, ex.edges)
    trees_r, f0_r = gbt_fit_ref(
        Xb[train], y[train].astype(np.float32), spec,
        sample_weight=w[train].astype(np.float32))
    assert abs(f0_d - f0_r) < 1e-5
    agree = 0
    for (fd, td, vd), (fr, tr, vr) in zip(trees_d, trees_r):
        if np.array_equal(fd, fr) and np.array_equal(td, tr):
            agree += 1
            np.testing.assert_allclose(vd, vr, rtol=1e-3, atol=1e-5)
    # weighted g/h float-atomic sums have more near-tied splits than the
    # unweighted integer-exact path: require a majority of identical
    # trees plus matching model QUALITY (near-tie flips partition
    # near-equal-loss regions, so accuracy must agree tightly)
    assert agree >= 2, f"only {agree}/6 weighted trees match"
    raw_d = gbt_predict_raw(Xb[train], trees_d, f0_d)
    raw_r = gbt_predict_raw(Xb[train], trees_r, f0_r)
    acc_d = float(np.mean((raw_d > 0) == (y[train] > 0.5)))
    acc_r = float(np.mean((raw_r > 0) == (y[train] > 0.5)))
    assert abs(acc_d - acc_r) < 0.02, (acc_d, acc_r)

    # unit weights reproduce the unweighted device model exactly
    (m1, _), _, _ = ex.fit(train, plan,
                           sample_weight=np.ones(X.shape[0]))
    (m0, _), _, _ = ex.fit(train, plan)
    for (fa, ta, va), (fb, tb, vb) in zip(m1[0][0], m0[0][0]):
        assert np.array_equal(fa, fb) and np.array_equal(ta, tb)


def test_device_scores_match_reference(problem):
    X, y = problem
    dev = torch.device("cuda", 0)
    ex = DeviceGBTExecutor(X, y.astype(np.float32), dev)
    spec = GBTSpec(n_estimators=15, max_depth=3, learning_rate=0.3)
    plan = _GBTPlan([spec])
    train = np.arange(15000, dtype=np.int64)
    test = np.arange(15000, 20000, dtype=np.int64)
    fitted, _, _ = ex.fit(train, plan)
    acc_d = ex.scores(test, fitted, plan, "accuracy")[0]

    models, edges = fitted
    trees, f0 = models[0]
    Xb = bin_data(X, edges)
    raw = gbt_predict_raw(Xb[test], trees, f0)
    acc_r = float(np.mean((raw > 0) == y[test]))
    assert abs(acc_d - acc_r) < 1e-6   # same trees -> same predictions
    assert acc_d > 0.8


def test_device_regression(problem):
    X, _ = problem
    rng = np.random.default_rng(2)
    yr = (2 * X[:, 0] + np.sin(X[:, 1]) +
          0.1 * rng.standard_normal(len(X))).astype(np.float32)
    dev = torch.device("cuda", 0)
    ex = DeviceGBTExecutor(X, yr, dev)
    spec = GBTSpec(loss=LOSS_SQUARED, n_estimators=30, max_depth=3,
                   learning_rate=0.2)
    plan = _GBTPlan([spec])
    train = np.arange(15000, dtype=np.int64)
    test = np.arange(15000, 20000, dtype=np.int64)
    fitted, _, _ = ex.fit(train, plan)
    r2 = ex.scores(test, fitted, plan, "r2")[0]
    assert r2 > 0.9


def test_device_grid_search_end_to_end(problem):
    X, y = problem
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import GradientBoostingClassifier

    gs = ska.GridSearchCV(
        None, GradientBoostingClassifier(n_estimators=15),
        {"max_depth": [2, 4], "learning_rate": [0.1, 0.3]}, cv=3,
        return_train_score=False).fit(X, y)
    assert len(gs.cv_results_["params"]) == 4
    assert gs.best_score_ > 0.8
    assert gs.predict(X[:50]).shape == (50,)


def test_device_multiclass_matches_reference(problem):
    """K=3 softmax boosting: device trees vs gbt_fit_ref_multi on the same
    binned matrix; scores must agree (tree structure may differ only via
    fp32 sum order in the softmax grads — accuracy must match closely)."""
    from spark_sklearn_amd.models._gbt_ref import (gbt_fit_ref_multi,
                                                   gbt_predict_raw_multi)

    rng = np.random.default_rng(11)
    n, f, K = 20000, 16, 3
    centers = rng.standard_normal((K, f)) * 2
    y = rng.integers(0, K, n)
    X = (centers[y] + rng.standard_normal((n, f))).astype(np.float32)

    dev = torch.device("cuda", 0)
    classes = np.arange(K)
    ex = DeviceGBTExecutor(X, y.astype(np.float32), dev, classes=classes)
    spec = GBTSpec(n_estimators=10, max_depth=3, learning_rate=0.3,
                   min_samples_leaf=20)
    plan = _GBTPlan([spec], K)
    train = np.arange(15000, dtype=np.int64)
    test = np.arange(15000, n, dtype=np.int64)
    fitted, n_iter, _ = ex.fit(train, plan)
    assert n_iter[0] == 10
    acc_d = ex.scores(test, fitted, plan, "accuracy")[0]

    models, edges = fitted
    Xb = bin_data(X, edges)
    trees_r, f0_r = gbt_fit_ref_multi(Xb[train], y[train].astype(np.int64),
                                      K, spec)
    raw = gbt_predict_raw_multi(Xb[test], trees_r, f0_r)
    acc_r = float(np.mean(raw.argmax(axis=1) == y[test]))
    assert acc_d > 0.9
    assert abs(acc_d - acc_r) < 0.01

    # downloaded host-format trees predict the same as device scoring
    trees_h, f0_h = models[0]
    raw_h = gbt_predict_raw_multi(Xb[test], trees_h, f0_h)
    acc_h = float(np.mean(raw_h.argmax(axis=1) == y[test]))
    assert abs(acc_h - acc_d) < 1e-6


def test_device_multiclass_grid_search(problem):
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import GradientBoostingClassifier

    rng = np.random.default_rng(5)
    n, f, K = 12000, 12, 4
    centers = rng.standard_normal((K, f)) * 2
    y = rng.integers(0, K, n)
    X = (centers[y] + rng.standard_normal((n, f))).astype(np.float32)
    gs = ska.GridSearchCV(
        None, GradientBoostingClassifier(n_estimators=12),
        {"max_depth": [2, 3]}, cv=3, return_train_score=False).fit(X, y)
    assert gs.best_score_ > 0.9
    assert gs.predict(X[:9]).shape == (9,)


def test_deterministic_mode(problem, monkeypatch):
    """SKA_DETERMINISTIC_GBT=1: fixed-point histogram/node-sum
    accumulation makes tree structures bit-stable run to run for
    classification losses, and the trees still match the reference."""
    import os

    from spark_sklearn_amd.models._gbt_ref import (GBTSpec, gbt_fit_ref)

    X, y = problem
    dev = torch.device("cuda", 0)
    spec = GBTSpec(n_estimators=6, max_depth=4, learning_rate=0.3,
                   min_samples_leaf=20)
    train = np.arange(15000, dtype=np.int64)

    monkeypatch.setenv("SKA_DETERMINISTIC_GBT", "1")
    outs = []
    for _ in range(2):
        ex = DeviceGBTExecutor(np.ascontiguousarray(X),
                               y.astype(np.float32), dev)
        (models, edges), _, _ = ex.fit(train, _GBTPlan([spec]))
        trees, f0 = models[0]
        outs.append([(tuple(f.tolist()), tuple(t.tolist()),
                      tuple(v.tolist())) for f, t, v in trees])
    assert outs[0] == outs[1]          # bit-identical, values included

    # and still structure-identical to the numpy reference
    Xb = bin_data(X, edges)
    trees_r, _ = gbt_fit_ref(Xb[train], y[train].astype(np.float32), spec)
    ref_struct = [(tuple(np.asarray(f).tolist()),
                   tuple(np.asarray(t).tolist())) for f, t, v in trees_r]
    dev_struct = [(f, t) for f, t, v in outs[0]]
    assert dev_struct == ref_struct


@pytest.mark.gpu
def test_gbt_depth8_device(problem):
    """Depth-8 boosting exercises the node-grouped histogram path."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import GradientBoostingClassifier

    X, y = problem
    gs = ska.GridSearchCV(None,
                          GradientBoostingClassifier(n_estimators=10,
                                                     max_depth=8,
                                                     min_samples_leaf=5),
                          {"learning_rate": [0.2]}, cv=2)
    gs.fit(X, y)
    assert gs.best_score_ > 0.85


@pytest.mark.gpu
def test_deterministic_mode_squared_loss(problem, monkeypatch):
    """Round-2: SKA_DETERMINISTIC_GBT=1 now covers squared loss via a
    per-round max-|g| reduction setting the fixed-point bias/scale."""
    from spark_sklearn_amd.models._gbt_ref import LOSS_SQUARED, GBTSpec

    X, y = problem
    dev = torch.device("cuda", 0)
    yr = (y.astype(np.float32) * 3.0 +
          X[:, 0].astype(np.float32))          # real-valued target
    spec = GBTSpec(loss=LOSS_SQUARED, n_estimators=5, max_depth=4,
                   learning_rate=0.3, min_samples_leaf=20)
    train = np.arange(15000, dtype=np.int64)

    monkeypatch.setenv("SKA_DETERMINISTIC_GBT", "1")
    outs = []
    for _ in range(2):
        ex = DeviceGBTExecutor(np.ascontiguousarray(X), yr, dev)
        (models, edges), _, _ = ex.fit(train, _GBTPlan([spec]))
        trees, f0 = models[0]
        outs.append([(tuple(f.tolist()), tuple(t.tolist()),
                      tuple(v.tolist())) for f, t, v in trees])
    assert outs[0] == outs[1]          # bit-identical incl. leaf values
