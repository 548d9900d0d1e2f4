"""GPU tests: device RandomForest kernels vs the numpy reference."""

import numpy as np
import pytest

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu

from sklearn.datasets import make_classification  # noqa: E402

from spark_sklearn_amd import GridSearchCV  # noqa: E402
from spark_sklearn_amd.models import RandomForestClassifier  # noqa: E402
from spark_sklearn_amd.models._gbt_ref import _tree_apply as _apply  # noqa: E402,E501
from spark_sklearn_amd.models._rf_ref import (RFSpec,  # noqa: E402
                                              rf_fit_ref, rf_predict_ref)


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


def _ext():
    from spark_sklearn_amd.ops import load_extension

    return load_extension()


def _dev_trees(ext, dev, Xb_np, y_np, spec, K=0, sample_weight=None):
    Xb = torch.as_tensor(Xb_np).to(dev)
    Xbt = Xb.t().contiguous()
    idx = torch.arange(Xb_np.shape[0], dtype=torch.int32, device=dev)
    y = torch.as_tensor(y_np.astype(np.float32)).to(dev)
    sw = (torch.as_tensor(np.asarray(sample_weight, np.float32)).to(dev)
          if sample_weight is not None else
          torch.empty(0, dtype=torch.float32, device=dev))
    feats, thrs, vals = ext.rf_fit(
        Xb, Xbt, idx, y, K, spec.n_estimators, spec.max_depth,
        spec.min_samples_leaf, spec.max_features, spec.random_state,
        1 if spec.bootstrap else 0,
        1 if spec.splitter == "random" else 0, sw)
    return (feats.cpu().numpy(), thrs.cpu().numpy().astype(np.uint8),
            vals.cpu().numpy())


def test_rf_device_weighted_matches_ref(dev):
    """Weighted gini on device (w-scaled planes + sum(w) h channel,
    classic-layout per-tree path) vs the weighted numpy reference;
    binary and multiclass."""
    rng = np.random.default_rng(3)
    n = 4000
    Xb = rng.integers(0, 64, size=(n, 16), dtype=np.uint8)
    y = (Xb[:, 0].astype(np.float32) + 0.5 * Xb[:, 3] > 40).astype(
        np.float32)
    w = rng.uniform(0.5, 2.0, size=n)
    spec = RFSpec(n_estimators=6, max_depth=3, min_samples_leaf=2,
                  max_features=8, random_state=11)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec, sample_weight=w)
    ref = rf_fit_ref(Xb, y, spec, sample_weight=w)
    agree = 0
    for t, (f_r, t_r, v_r) in enumerate(ref):
        if np.array_equal(f_d[t], f_r) and np.array_equal(t_d[t], t_r):
            agree += 1
            np.testing.assert_allclose(v_d[t], v_r, rtol=2e-3, atol=1e-5)
    # the device h channel is fixed-point-quantized (hscale), so rare
    # near-tied splits may flip vs the fp64 reference; require a
    # majority plus matching forest QUALITY below
    assert agree >= 2, f"only {agree}/6 weighted trees match"
    pred_d = sum(_apply(Xb, f_d[t], t_d[t], v_d[t])
                 for t in range(6))
    ref_flat = [ref[t] for t in range(6)]
    pred_r = sum(_apply(Xb, fr, tr, vr) for fr, tr, vr in ref_flat)
    acc_d = float(np.mean((pred_d > 0.5 * 6 / 6) == (y > 0.5)))
    acc_r = float(np.mean((pred_r > 0.5 * 6 / 6) == (y > 0.5)))
    assert abs(acc_d - acc_r) < 0.02, (acc_d, acc_r)

    ym = (Xb[:, 1] // 22).astype(np.float32)       # 3 classes
    K = 3
    spec_m = RFSpec(n_estimators=4, max_depth=3, min_samples_leaf=2,
                    max_features=8, random_state=5, n_classes=K)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, ym, spec_m, K=K,
                               sample_weight=w)
    ref = rf_fit_ref(Xb, ym, spec_m, sample_weight=w)
    agree = 0
    for t, planes in enumerate(ref):
        f_r = planes[0][0]
        if np.array_equal(f_d[t * K], f_r):
            agree += 1
    assert agree >= 1, f"only {agree}/4 weighted multiclass trees match"

    # unit weights reproduce the unweighted forest structures
    f_1, t_1, _ = _dev_trees(_ext(), dev, Xb, y, spec,
                             sample_weight=np.ones(n))
    f_0, t_0, _ = _dev_trees(_ext(), dev, Xb, y, spec)
    assert np.array_equal(f_1, f_0) and np.array_equal(t_1, t_0)


def test_rf_device_matches_ref_binary(dev):
    rng = np.random.default_rng(0)
    n = 4000
    Xb = rng.integers(0, 64, size=(n, 16), dtype=np.uint8)
    # additive, well-separated target: every candidate split has a clear
    # margin, so fp32 (device) vs fp64 (numpy) gain rounding cannot flip
    # the argmax (an XOR target makes all first-level gains ~0 and is
    # decided by noise)
    y = (Xb[:, 0].astype(np.float32) + 0.5 * Xb[:, 3] > 40).astype(
        np.float32)
    # exact structure at depth 3 (deeper trees produce small discrete
    # nodes where genuinely-tied gains are decided by fp32-vs-fp64
    # rounding; depth>=4 is covered statistically below)
    spec = RFSpec(n_estimators=6, max_depth=3, min_samples_leaf=2,
                  max_features=8, random_state=11)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec)
    ref = rf_fit_ref(Xb, y, spec)
    for t, (f_r, t_r, v_r) in enumerate(ref):
        assert np.array_equal(f_d[t], f_r), f"tree {t} feats differ"
        split = f_d[t] >= 0
        assert np.array_equal(t_d[t][split], t_r[split]), f"tree {t} thrs"
        np.testing.assert_allclose(v_d[t], v_r, atol=2e-5)


def test_rf_device_deep_trees_statistical_parity(dev):
    rng = np.random.default_rng(4)
    n = 6000
    Xb = rng.integers(0, 64, size=(n, 16), dtype=np.uint8)
    y = (Xb[:, 0].astype(np.float32) + 0.5 * Xb[:, 3] +
         0.25 * Xb[:, 7] > 46).astype(np.float32)
    spec = RFSpec(n_estimators=10, max_depth=6, min_samples_leaf=2,
                  max_features=8, random_state=2)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec)
    ref = rf_fit_ref(Xb, y, spec)
    trees_d = [(f_d[t], t_d[t], v_d[t]) for t in range(len(ref))]
    p_dev = rf_predict_ref(Xb, trees_d)
    p_ref = rf_predict_ref(Xb, ref)
    acc_d = float(np.mean((p_dev > 0.5) == (y > 0.5)))
    acc_r = float(np.mean((p_ref > 0.5) == (y > 0.5)))
    assert acc_d > 0.95
    assert abs(acc_d - acc_r) < 0.01
    # most nodes agree even where ties exist
    agree = np.mean([np.mean(f_d[t] == r[0]) for t, r in enumerate(ref)])
    assert agree > 0.8


def test_rf_device_matches_ref_multiclass(dev):
    rng = np.random.default_rng(2)
    n, K = 3000, 5
    Xb = rng.integers(0, 32, size=(n, 12), dtype=np.uint8)
    y = (Xb[:, 1] // 7 + (Xb[:, 5] > 16)).astype(np.float32) % K
    spec = RFSpec(n_estimators=4, max_depth=3, min_samples_leaf=2,
                  max_features=6, random_state=5, n_classes=K)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec, K=K)
    ref = rf_fit_ref(Xb, y, spec)
    for t, planes in enumerate(ref):
        for c, (f_r, t_r, v_r) in enumerate(planes):
            i = t * K + c
            assert np.array_equal(f_d[i], f_r), f"tree {t} class {c}"
            split = f_d[i] >= 0
            assert np.array_equal(t_d[i][split], t_r[split])
            np.testing.assert_allclose(v_d[i], v_r, atol=2e-5)


def test_rf_device_no_bootstrap_no_mask_matches(dev):
    rng = np.random.default_rng(3)
    Xb = rng.integers(0, 64, size=(2000, 8), dtype=np.uint8)
    y = (Xb[:, 2] > 40).astype(np.float32)
    spec = RFSpec(n_estimators=2, max_depth=3, bootstrap=False,
                  max_features=0, random_state=0)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec)
    ref = rf_fit_ref(Xb, y, spec)
    for t, (f_r, t_r, v_r) in enumerate(ref):
        assert np.array_equal(f_d[t], f_r)
        np.testing.assert_allclose(v_d[t], v_r, atol=2e-5)


def test_rf_grid_search_device_end_to_end(dev):
    X, y = make_classification(n_samples=20_000, n_features=32,
                               n_informative=12, random_state=0)
    gs = GridSearchCV(None, RandomForestClassifier(random_state=0),
                      {"max_depth": [4, 6], "n_estimators": [10, 25]},
                      cv=3)
    gs.fit(X, y)
    assert gs.best_score_ > 0.85
    assert gs.cv_results_["mean_test_score"].shape == (4,)
    # device executor actually ran
    from spark_sklearn_amd.engine import get_family

    fam = get_family(RandomForestClassifier())
    assert fam.name == "rf"


def test_rf_device_digits_shape_multiclass_accuracy(dev):
    from sklearn.datasets import load_digits

    X, y = load_digits(return_X_y=True)
    gs = GridSearchCV(None, RandomForestClassifier(random_state=0),
                      {"max_depth": [6], "n_estimators": [20, 40]}, cv=3)
    gs.fit(X, y)
    assert gs.best_score_ > 0.85


def test_rf_deep_trees_depth8(dev):
    """Depth-8 forests exercise the node-grouped histogram path (levels
    whose histograms exceed one block's LDS)."""
    rng = np.random.default_rng(7)
    n = 20_000
    Xb = rng.integers(0, 64, size=(n, 16), dtype=np.uint8)
    y = (Xb[:, 0].astype(np.float32) + 0.5 * Xb[:, 3] +
         0.25 * Xb[:, 7] + 0.125 * Xb[:, 11] > 52).astype(np.float32)
    spec = RFSpec(n_estimators=8, max_depth=8, min_samples_leaf=2,
                  max_features=8, random_state=1)
    f_d, t_d, v_d = _dev_trees(_ext(), dev, Xb, y, spec)
    ref = rf_fit_ref(Xb, y, spec)
    trees_d = [(f_d[t], t_d[t], v_d[t]) for t in range(len(ref))]
    p_dev = rf_predict_ref(Xb, trees_d)
    p_ref = rf_predict_ref(Xb, ref)
    acc_d = float(np.mean((p_dev > 0.5) == (y > 0.5)))
    acc_r = float(np.mean((p_ref > 0.5) == (y > 0.5)))
    assert acc_d > 0.97
    assert abs(acc_d - acc_r) < 0.01


def test_extratrees_device_matches_ref(dev):
    rng = np.random.default_rng(9)
    n = 4000
    Xb = rng.integers(0, 64, size=(n, 12), dtype=np.uint8)
    y = (Xb[:, 0].astype(np.float32) + 0.5 * Xb[:, 4] > 40).astype(
        np.float32)
    spec = RFSpec(n_estimators=5, max_depth=3, min_samples_leaf=2,
                  max_features=6, bootstrap=False, random_state=4,
                  splitter="random")
    ext = _ext()
    Xb_t = torch.as_tensor(Xb).to(dev)
    idx = torch.arange(n, dtype=torch.int32, device=dev)
    yt = torch.as_tensor(y).to(dev)
    feats, thrs, vals = ext.rf_fit(
        Xb_t, Xb_t.t().contiguous(), idx, yt, 0, 5, 3, 2, 6, 4, 0, 1,
        torch.empty(0, dtype=torch.float32, device=dev))
    ref = rf_fit_ref(Xb, y, spec)
    f_d = feats.cpu().numpy()
    t_d = thrs.cpu().numpy().astype(np.uint8)
    v_d = vals.cpu().numpy()
    for t, (f_r, t_r, v_r) in enumerate(ref):
        assert np.array_equal(f_d[t], f_r), f"tree {t}"
        split = f_d[t] >= 0
        assert np.array_equal(t_d[t][split], t_r[split])
        np.testing.assert_allclose(v_d[t], v_r, atol=2e-5)


def test_extratrees_multiclass_device_matches_ref(dev):
    """Random-threshold splitter through the K-class gini finder."""
    rng = np.random.default_rng(12)
    n, K = 3000, 4
    Xb = rng.integers(0, 32, size=(n, 10), dtype=np.uint8)
    y = ((Xb[:, 2] // 8) % K).astype(np.float32)
    spec = RFSpec(n_estimators=4, max_depth=3, min_samples_leaf=2,
                  max_features=5, bootstrap=False, random_state=6,
                  splitter="random", n_classes=K)
    ext = _ext()
    Xb_t = torch.as_tensor(Xb).to(dev)
    idx = torch.arange(n, dtype=torch.int32, device=dev)
    yt = torch.as_tensor(y).to(dev)
    feats, thrs, vals = ext.rf_fit(
        Xb_t, Xb_t.t().contiguous(), idx, yt, K, 4, 3, 2, 5, 6, 0, 1,
        torch.empty(0, dtype=torch.float32, device=dev))
    ref = rf_fit_ref(Xb, y, spec)
    f_d = feats.cpu().numpy()
    t_d = thrs.cpu().numpy().astype(np.uint8)
    v_d = vals.cpu().numpy()
    for t, planes in enumerate(ref):
        for c, (f_r, t_r, v_r) in enumerate(planes):
            i = t * K + c
            assert np.array_equal(f_d[i], f_r), f"tree {t} class {c}"
            split = f_d[i] >= 0
            assert np.array_equal(t_d[i][split], t_r[split])
            np.testing.assert_allclose(v_d[i], v_r, atol=2e-5)
