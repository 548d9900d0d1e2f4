"""Linear-model Converter: device-native models <-> genuine sklearn objects.

Behavioral counterpart of the reference's ``Converter`` (``converter.py``
~L20-120 [R]), which copies ``coefficients``/``intercept``
between pyspark.ml linear models and sklearn ones.  Here the "other side"
is the MI355X-native estimator families of :mod:`spark_sklearn_amd.models`
(device-resident fitted parameters): the Converter produces genuine sklearn
estimator objects from fitted device models -- so ``best_estimator_`` is
always usable with the whole sklearn ecosystem -- and imports sklearn linear
models onto the device path.

Supported pairs (same coverage as the reference: logistic + linear
regression; ``ValueError`` on anything else):

* ``models.LogisticRegression``  <->  ``sklearn.linear_model.LogisticRegression``
* ``models.LinearRegression``    <->  ``sklearn.linear_model.LinearRegression``
* ``models.LinearSVC``           <->  ``sklearn.svm.LinearSVC``
"""

from __future__ import annotations

import numpy as np

__all__ = ["Converter"]


class Converter:
    """Convert fitted models between the device-native and sklearn worlds."""

    def __init__(self, ctx=None):
        self.ctx = ctx  # kept for signature parity with Converter(sc) [R]

    # -- device -> sklearn (the reference's toSKLearn direction) ----------
    def toSKLearn(self, model):
        from sklearn.linear_model import LinearRegression as SKLinReg
        from sklearn.linear_model import LogisticRegression as SKLogReg
        from sklearn.svm import LinearSVC as SKLinearSVC

        from .models import LinearRegression, LinearSVC, LogisticRegression

        if isinstance(model, LogisticRegression):
            sk = SKLogReg(C=model.C, penalty=model.penalty,
                          fit_intercept=model.fit_intercept)
            _install_fitted(sk, _check_fitted_src(model),
                            classes=model.classes_)
            return sk
        if isinstance(model, LinearSVC):
            sk = SKLinearSVC(C=model.C, fit_intercept=model.fit_intercept)
            _install_fitted(sk, _check_fitted_src(model),
                            classes=model.classes_)
            return sk
        if isinstance(model, LinearRegression):
            sk = SKLinReg(fit_intercept=model.fit_intercept)
            _install_fitted(sk, model)
            return sk
        raise ValueError(
            f"Converter.toSKLearn does not support {type(model).__name__}; "
            f"supported: LogisticRegression, LinearSVC, LinearRegression")

    # -- sklearn -> device (the reference's toSpark direction) ------------
    def toHip(self, model):
        from sklearn.linear_model import LinearRegression as SKLinReg
        from sklearn.linear_model import LogisticRegression as SKLogReg
        from sklearn.svm import LinearSVC as SKLinearSVC

        from .models import LinearRegression, LinearSVC, LogisticRegression

        if isinstance(model, SKLogReg):
            out = LogisticRegression(C=model.C,
                                     fit_intercept=model.fit_intercept)
            _install_fitted(out, _check_fitted_src(model),
                            classes=model.classes_)
            return out
        if isinstance(model, SKLinearSVC):
            out = LinearSVC(C=model.C, fit_intercept=model.fit_intercept)
            _install_fitted(out, _check_fitted_src(model),
                            classes=model.classes_)
            return out
        if isinstance(model, SKLinReg):
            out = LinearRegression(fit_intercept=model.fit_intercept)
            _install_fitted(out, model)
            return out
        raise ValueError(
            f"Converter.toHip does not support {type(model).__name__}; "
            f"supported: sklearn LogisticRegression, LinearSVC, "
            f"LinearRegression")

    # name kept as an alias so reference-era call sites read naturally
    toSpark = toHip


def _check_fitted_src(model):
    if not hasattr(model, "coef_"):
        raise ValueError(f"{type(model).__name__} instance is not fitted")
    return model


def _install_fitted(dst, src, classes=None):
    """Copy the fitted linear parameters (coef_, intercept_) src -> dst."""
    if not hasattr(src, "coef_"):
        raise ValueError(f"{type(src).__name__} instance is not fitted")
    dst.coef_ = np.array(src.coef_, dtype=np.float64, copy=True)
    intercept = getattr(src, "intercept_", 0.0)
    dst.intercept_ = (np.array(intercept, dtype=np.float64, copy=True)
                      if np.ndim(intercept) else float(intercept))
    if classes is not None:
        dst.classes_ = np.asarray(classes)
    n_features = dst.coef_.shape[-1]
    dst.n_features_in_ = int(n_features)
    if hasattr(src, "n_iter_"):
        dst.n_iter_ = np.asarray(src.n_iter_)
