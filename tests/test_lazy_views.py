"""Contract tests for the on-demand host-model views the device
executors return (install_fitted / host scoring index them like lists;
materialization pulls device planes at most once per slot)."""

from dataclasses import dataclass

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from spark_sklearn_amd.ops.gbt_executor import _LazyModels  # noqa: E402
from spark_sklearn_amd.ops.rf_executor import _LazyForests  # noqa: E402


@dataclass
class _Spec:
    n_estimators: int


def _planes(T, K, NN, seed=0):
    rng = np.random.default_rng(seed)
    feats = torch.as_tensor(
        rng.integers(-1, 5, size=(T * max(K, 1), NN)).astype(np.int32))
    thrs = torch.as_tensor(
        rng.integers(0, 255, size=(T * max(K, 1), NN)).astype(np.int32))
    vals = torch.as_tensor(
        rng.standard_normal((T * max(K, 1), NN)).astype(np.float32))
    return feats, thrs, vals


def test_lazy_forests_binary_and_multi():
    T, NN = 3, 7
    f, t, v = _planes(T, 1, NN)
    models_dev = [("flat", f, t, v, _Spec(T)), None]
    view = _LazyForests(models_dev, K=2)
    assert len(view) == 2
    trees = view[0]
    assert len(trees) == T
    ft, tt, vt = trees[1]
    np.testing.assert_array_equal(ft, f.numpy()[1])
    assert tt.dtype == np.uint8
    assert view[1] == []            # skipped slot
    assert view[0] is trees         # cached

    K = 3
    f, t, v = _planes(T, K, NN, seed=1)
    mv = _LazyForests([("multi", f, t, v, _Spec(T))], K=K)
    planes = mv[0]
    assert len(planes) == T and len(planes[0]) == K
    np.testing.assert_array_equal(planes[2][1][0], f.numpy()[2 * K + 1])


def test_lazy_gbt_models():
    T, NN = 4, 7
    f, t, v = _planes(T, 1, NN, seed=2)
    view = _LazyModels([(f, t, v, -0.25, 0), None], K=0)
    trees, f0 = view[0]
    assert len(trees) == T and f0 == -0.25
    assert view[1] == ([], 0.0)

    K = 3
    f, t, v = _planes(T, K, NN, seed=3)
    f0 = np.zeros(K, np.float32)
    mv = _LazyModels([(f, t, v, f0, _Spec(T))], K=K)
    trees, f0_out = mv[0]
    assert len(trees) == T and len(trees[0]) == K
    np.testing.assert_array_equal(trees[1][2][0], f.numpy()[1 * K + 2])
