"""MI355X-native histogram gradient boosting.

sklearn-compatible GradientBoostingClassifier / GradientBoostingRegressor
whose inner loop is the K3 HIP kernel family (per-(node, feature) gradient/
hessian histograms in LDS, best-split scan, partition, leaf update --
SURVEY.md section 2.5 K3) on GPU and models/_gbt_ref.py on CPU.

Implementation is HISTOGRAM-based boosting (the modern sklearn
HistGradientBoosting* algorithm; the classic exact-split surface is kept
for reference-era naming parity).  Binary classification (logistic loss)
and regression (squared loss); quantile binning to <=256 uint8 bins once
per dataset.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from sklearn.base import BaseEstimator, ClassifierMixin, RegressorMixin

from ._gbt_ref import (GBTSpec, LOSS_LOGISTIC, LOSS_SQUARED, bin_data,
                       gbt_fit_ref, gbt_fit_ref_multi, gbt_predict_raw,
                       gbt_predict_raw_multi, quantile_bins, sigmoid,
                       softmax_rows)

__all__ = ["GradientBoostingClassifier", "GradientBoostingRegressor"]


class _GBTBase(BaseEstimator):
    _hip_batched = True
    _engine_family = "gbt"
    _loss_id = LOSS_LOGISTIC

    def __init__(self, n_estimators=100, learning_rate=0.1, max_depth=3,
                 min_samples_leaf=20, l2_regularization=0.0, max_bins=256,
                 random_state=None):
        self.n_estimators = n_estimators
        self.learning_rate = learning_rate
        self.max_depth = max_depth
        self.min_samples_leaf = min_samples_leaf
        self.l2_regularization = l2_regularization
        self.max_bins = max_bins
        self.random_state = random_state

    def _spec(self) -> GBTSpec:
        if not (1 <= int(self.max_depth) <= 10):
            raise ValueError("max_depth must be in [1, 10]")
        if int(self.n_estimators) < 1:
            raise ValueError("n_estimators must be >= 1")
        if not (2 <= int(self.max_bins) <= 256):
            raise ValueError("max_bins must be in [2, 256]")
        return GBTSpec(loss=self._loss_id,
                       n_estimators=int(self.n_estimators),
                       learning_rate=float(self.learning_rate),
                       max_depth=int(self.max_depth),
                       min_samples_leaf=int(self.min_samples_leaf),
                       l2_regularization=float(self.l2_regularization),
                       max_bins=int(self.max_bins))

    def fit(self, X, y, sample_weight=None):
        from ..engine import fit_single

        return fit_single(self, X, y, sample_weight=sample_weight)

    @property
    def feature_importances_(self):
        """Split-frequency importances, normalized to sum to 1 (documented
        divergence: sklearn weighs splits by impurity decrease; the device
        trees store (feature, threshold, value) only, so frequency is the
        faithful statistic -- same ranking on well-separated problems)."""
        self._check_fitted()
        counts = np.zeros(self.n_features_in_, dtype=np.float64)
        for tree in self._trees:
            groups = tree if isinstance(tree, list) else [tree]
            for feats, thrs, vals in groups:
                for f in np.asarray(feats).ravel():
                    if f >= 0:
                        counts[int(f)] += 1.0
        tot = counts.sum()
        return counts / tot if tot > 0 else counts

    def _raw_predict(self, X):
        self._check_fitted()
        Xb = bin_data(np.asarray(X), self._bin_edges)
        if np.ndim(self._f0) == 1:          # multiclass: [n, K]
            return gbt_predict_raw_multi(Xb, self._trees, self._f0)
        return gbt_predict_raw(Xb, self._trees, self._f0)

    def _check_fitted(self):
        if not hasattr(self, "_trees"):
            raise AttributeError(
                f"this {type(self).__name__} instance is not fitted yet")


class GradientBoostingClassifier(ClassifierMixin, _GBTBase):
    """Histogram gradient boosting classifier on CDNA4 (binary logistic;
    multiclass via softmax boosting: one tree per class per round)."""

    _loss_id = LOSS_LOGISTIC

    def predict_proba(self, X):
        raw = self._raw_predict(X)
        if raw.ndim == 2:
            return softmax_rows(raw)
        p = sigmoid(raw)
        return np.column_stack([1.0 - p, p])

    def predict(self, X):
        raw = self._raw_predict(X)
        if raw.ndim == 2:
            return self.classes_[np.argmax(raw, axis=1)]
        return self.classes_[(raw > 0).astype(int)]

    def decision_function(self, X):
        return self._raw_predict(X)


class GradientBoostingRegressor(RegressorMixin, _GBTBase):
    """Histogram gradient boosting for regression (squared loss)."""

    _loss_id = LOSS_SQUARED

    def predict(self, X):
        return self._raw_predict(X).astype(np.float64)


class GBTFamily:
    """Engine adapter: sequential per-candidate boosting over shared bins.

    Unlike the linear/kmeans families there is no single-kernel candidate
    batching (tree shapes diverge per candidate); the shared work is the
    one-time quantile binning + resident uint8 matrix, and candidates of a
    chunk run back-to-back on the device without host round-trips between
    boosting iterations.
    """

    name = "gbt"

    @staticmethod
    def encode_targets(estimator, y):
        from sklearn.base import is_classifier

        y = np.asarray(y)
        if is_classifier(estimator):
            classes, y_enc = np.unique(y, return_inverse=True)
            if len(classes) > 64:
                raise ValueError("at most 64 classes supported")
            return classes, y_enc.astype(np.float32)
        return None, np.asarray(y, dtype=np.float32)

    @staticmethod
    def cols_per_cand(n_classes):
        return 1

    @staticmethod
    def cand_chunk(estimator, params_list, column_batch):
        return 32     # loop granularity only (no cross-candidate kernel)

    @staticmethod
    def make_executor(X, y_enc, classes, device):
        if device is not None:
            from ..ops.gbt_executor import DeviceGBTExecutor

            return DeviceGBTExecutor(X, y_enc, device, classes=classes)
        return _HostGBTExecutor(X, y_enc, classes)

    @staticmethod
    def build_plan(estimator, params_list, n_samples, executor, n_classes,
                   error_score):
        from sklearn.base import clone

        proto = clone(estimator)
        base_params = proto.get_params()
        specs, bad = [], {}
        for slot, params in enumerate(params_list):
            try:
                unknown = set(params) - set(base_params)
                if unknown:
                    raise ValueError(
                        f"invalid parameters {sorted(unknown)} for "
                        f"{type(proto).__name__}")
                proto.__dict__.update(base_params)
                proto.__dict__.update(params)
                specs.append(proto._spec())
            except Exception as exc:  # noqa: BLE001 - error_score contract
                if error_score == "raise":
                    raise
                bad[slot] = exc
                specs.append(GBTSpec(n_estimators=0, max_depth=1))
        plan = _GBTPlan(specs, n_classes)
        return plan, bad

    @staticmethod
    def metric_for(scorer, estimator) -> Optional[str]:
        from sklearn.base import is_classifier

        if scorer is None or type(scorer).__name__ == "_PassthroughScorer":
            return "accuracy" if is_classifier(estimator) else "r2"
        fn = getattr(scorer, "_score_func", None)
        name = getattr(fn, "__name__", "")
        if name == "accuracy_score":
            return "accuracy"
        if name == "r2_score":
            return "r2"
        return None

    @staticmethod
    def install_fitted(est, fitted, slot, plan, classes):
        models, edges = fitted
        trees, f0 = models[slot]
        est._trees = trees
        est._f0 = f0
        est._bin_edges = edges
        if classes is not None:
            est.classes_ = np.asarray(classes)
        est.n_features_in_ = len(edges)
        est.n_iter_ = len(trees)
        return est


class _GBTPlan:
    def __init__(self, specs, n_label_classes=0):
        self.specs = specs
        self.col_to_cand = np.arange(len(specs), dtype=np.int32)
        self.n_classes = 1            # engine chunking unit (per candidate)
        self.n_label_classes = n_label_classes

    @property
    def B(self):
        return len(self.specs)


class _HostGBTExecutor:
    """Numpy execution (reference semantics); bins built once per matrix."""

    def __init__(self, X, y_enc, classes=None):
        X = np.asarray(X)
        self.edges = quantile_bins(X)
        self.Xb = bin_data(X, self.edges)
        self.y = y_enc
        self.K = len(classes) if classes is not None else 0
        self.msq = 0.0   # unused by this family

    supports_sample_weight = True

    def fit(self, train_idx, plan, sample_weight=None):
        Xb_tr = self.Xb[train_idx]
        y_tr = self.y[train_idx]
        sw = (np.asarray(sample_weight, np.float32)[train_idx]
              if sample_weight is not None else None)
        models = []
        n_iter = np.empty(plan.B, np.int32)
        for i, spec in enumerate(plan.specs):
            if self.K > 2:
                trees, f0 = gbt_fit_ref_multi(
                    Xb_tr, y_tr.astype(np.int64), self.K, spec,
                    sample_weight=sw)
            else:
                trees, f0 = gbt_fit_ref(Xb_tr, y_tr, spec,
                                        sample_weight=sw)
            models.append((trees, f0))
            n_iter[i] = len(trees)
        return (models, self.edges), n_iter, None

    def scores(self, idx, fitted, plan, metric):
        models, _ = fitted
        Xb = self.Xb[idx]
        yv = self.y[idx]
        out = np.empty(len(models))
        for i, (trees, f0) in enumerate(models):
            if self.K > 2:
                raw = gbt_predict_raw_multi(Xb, trees, f0)
                assert metric == "accuracy"
                out[i] = float(np.mean(raw.argmax(axis=1) ==
                                       yv.astype(np.int64)))
                continue
            raw = gbt_predict_raw(Xb, trees, f0)
            if metric == "accuracy":
                out[i] = float(np.mean((raw > 0) == (yv > 0.5)))
            elif metric == "r2":
                ss_tot = float(np.sum((yv - yv.mean()) ** 2))
                out[i] = 1.0 - float(np.sum((yv - raw) ** 2)) / \
                    max(ss_tot, 1e-12)
            else:
                raise ValueError(metric)
        return out
