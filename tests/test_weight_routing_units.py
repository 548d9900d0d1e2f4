"""Unit tests for the engine's weight-routing helpers."""

import numpy as np
import pytest

from spark_sklearn_amd.engine import (_class_weight_rows, _fold_weights,
                                      _route_fit_params)


class _NoWeightExec:
    pass


class _WeightExec:
    supports_sample_weight = True


def test_route_ignores_on_unsupported_executor():
    X = np.zeros((10, 2))
    w, scale, ignored = _route_fit_params(
        {"sample_weight": np.ones(10)}, _NoWeightExec(), X)
    assert w is None and scale == 1.0
    assert "sample_weight" in ignored


def test_route_normalizes_to_mean_one():
    X = np.zeros((8, 2))
    w, scale, ignored = _route_fit_params(
        {"sample_weight": np.full(8, 13.0)}, _WeightExec(), X)
    np.testing.assert_allclose(w, 1.0)
    assert scale == 1.0 and ignored == {}


def test_class_weight_rows_balanced_uses_train_fold():
    classes = np.array([0, 1])
    y = np.array([0, 0, 0, 1, 1, 1])
    # train fold sees 3:1 imbalance -> class-1 rows weigh 2x class-0
    train = np.array([0, 1, 2, 3])
    rows = _class_weight_rows("balanced", classes, y, train)
    np.testing.assert_allclose(rows[:3], 4 / (2 * 3))
    np.testing.assert_allclose(rows[3:], 4 / (2 * 1))


def test_fold_weights_zero_mass_raises():
    classes = np.array([0, 1])
    y = np.array([0, 0, 1, 1])
    train = np.array([0, 1])            # only class 0 in the fold
    with pytest.raises(ValueError, match="zeroes out"):
        _fold_weights(None, 1.0, {0: 0.0}, classes, y, train)


def test_fold_weights_combines_multiplicatively():
    classes = np.array([0, 1])
    y = np.array([0, 1, 0, 1])
    sw = np.array([1.0, 1.0, 3.0, 3.0], np.float32)
    train = np.arange(4)
    eff, mx = _fold_weights(sw, float(sw.max()), {0: 1.0, 1: 2.0},
                            classes, y, train)
    # ratios preserved: (1, 2, 3, 6) scaled to mean 1 over the fold
    np.testing.assert_allclose(eff / eff[0], [1, 2, 3, 6], rtol=1e-6)
    np.testing.assert_allclose(eff.mean(), 1.0, rtol=1e-6)
    assert mx == pytest.approx(eff.max())
