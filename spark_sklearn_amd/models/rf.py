"""MI355X-native RandomForest (bagged histogram trees, K3b kernels).

sklearn-compatible RandomForestClassifier / RandomForestRegressor whose
inner loop shares the K3 histogram machinery with the GBT family: one-time
256-bin quantile binning, per-level LDS histograms with sibling
subtraction, a gini/MSE split scan with exact-k per-node feature
subsampling, and a stateless-hash bootstrap resample per tree
(ops/csrc/ska_ops.hip::rf_fit).  CPU reference: models/_rf_ref.py.

This is the reference's motivating workload (the spark-sklearn README's
digits + RandomForest grid -- SURVEY.md section 6); spark-sklearn itself
shipped sklearn's RandomForest to Spark executors, so the
MI355X equivalent is a hand-written device forest over the resident
binned matrix.

Documented divergences from sklearn (see _rf_ref module docstring):
binned splits, hash-stream feature subsampling, max_depth required on the
device path (cap 10, node-grouped deep-level histograms).
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from sklearn.base import BaseEstimator, ClassifierMixin, RegressorMixin

from ._gbt_ref import bin_data, quantile_bins
from ._rf_ref import RFSpec, rf_fit_ref, rf_predict_ref

__all__ = ["RandomForestClassifier", "RandomForestRegressor",
           "ExtraTreesClassifier", "ExtraTreesRegressor"]

_DEVICE_MAX_DEPTH = 10


class _RFBase(BaseEstimator):
    _hip_batched = True
    _engine_family = "rf"
    _splitter = "best"

    def __init__(self, n_estimators=100, max_depth=10, min_samples_leaf=1,
                 max_features="sqrt", bootstrap=True, random_state=None):
        self.n_estimators = n_estimators
        self.max_depth = max_depth
        self.min_samples_leaf = min_samples_leaf
        self.max_features = max_features
        self.bootstrap = bootstrap
        self.random_state = random_state

    def _resolve_max_features(self, F):
        mf = self.max_features
        if mf is None:
            return 0
        if mf == "sqrt":
            return max(1, int(np.sqrt(F)))
        if mf == "log2":
            return max(1, int(np.log2(F)))
        if isinstance(mf, float):
            return max(1, int(mf * F))
        k = int(mf)
        if k < 1:
            raise ValueError("max_features must be >= 1")
        return k

    def _spec(self, F=None, n_classes=0) -> RFSpec:
        if self.max_depth is None:
            raise ValueError(
                "device RandomForest requires max_depth (<= %d); "
                "unlimited-depth forests are a host-path feature"
                % _DEVICE_MAX_DEPTH)
        if not (1 <= int(self.max_depth) <= _DEVICE_MAX_DEPTH):
            raise ValueError(
                f"max_depth must be in [1, {_DEVICE_MAX_DEPTH}]")
        if int(self.n_estimators) < 1:
            raise ValueError("n_estimators must be >= 1")
        return RFSpec(
            n_estimators=int(self.n_estimators),
            max_depth=int(self.max_depth),
            min_samples_leaf=int(self.min_samples_leaf),
            max_features=self._resolve_max_features(F) if F else 0,
            bootstrap=bool(self.bootstrap),
            random_state=int(self.random_state or 0),
            n_classes=n_classes,
            splitter=self._splitter,
        )

    def fit(self, X, y, sample_weight=None):
        from ..engine import fit_single

        fit_single(self, X, y, sample_weight=sample_weight)
        return self

    @property
    def feature_importances_(self):
        self._check_fitted()
        F = self.n_features_in_
        imp = np.zeros(F)
        for tree in self._trees:
            planes = tree if isinstance(tree, list) else [tree]
            feat = planes[0][0]
            for f in feat:
                if f >= 0:
                    imp[f] += 1.0
        s = imp.sum()
        return imp / s if s > 0 else imp

    def _forest_raw(self, X):
        self._check_fitted()
        X = np.asarray(X)
        Xb = bin_data(X, self._bin_edges)
        K = len(self.classes_) if hasattr(self, "classes_") else 0
        return rf_predict_ref(Xb, self._trees, K if K > 2 else 0)

    def _check_fitted(self):
        if not hasattr(self, "_trees"):
            raise RuntimeError("estimator is not fitted")


class RandomForestClassifier(ClassifierMixin, _RFBase):
    """Bagged histogram-tree classifier (gini splits; binary + K-class)."""

    def __init__(self, n_estimators=100, max_depth=10, min_samples_leaf=1,
                 max_features="sqrt", bootstrap=True, random_state=None,
                 class_weight=None):
        super().__init__(n_estimators=n_estimators, max_depth=max_depth,
                         min_samples_leaf=min_samples_leaf,
                         max_features=max_features, bootstrap=bootstrap,
                         random_state=random_state)
        self.class_weight = class_weight

    def predict_proba(self, X):
        raw = self._forest_raw(X)
        if raw.ndim == 2:                      # K > 2: per-class probs
            s = raw.sum(axis=1, keepdims=True)
            return raw / np.maximum(s, 1e-12)
        p1 = np.clip(raw, 0.0, 1.0)
        return np.stack([1.0 - p1, p1], axis=1)

    def predict(self, X):
        raw = self._forest_raw(X)
        if raw.ndim == 2:
            return self.classes_[raw.argmax(axis=1)]
        return self.classes_[(raw > 0.5).astype(np.int64)]


class ExtraTreesClassifier(RandomForestClassifier):
    """Extremely-randomized trees: one uniformly-drawn threshold bin per
    (tree, node, feature) instead of the best-gain scan; no bootstrap by
    default (sklearn semantics on the shared binned-forest machinery)."""

    _splitter = "random"

    def __init__(self, n_estimators=100, max_depth=10, min_samples_leaf=1,
                 max_features="sqrt", bootstrap=False, random_state=None,
                 class_weight=None):
        super().__init__(n_estimators=n_estimators, max_depth=max_depth,
                         min_samples_leaf=min_samples_leaf,
                         max_features=max_features, bootstrap=bootstrap,
                         random_state=random_state)
        self.class_weight = class_weight


class RandomForestRegressor(RegressorMixin, _RFBase):
    """Bagged histogram-tree regressor (MSE splits)."""

    def __init__(self, n_estimators=100, max_depth=10, min_samples_leaf=1,
                 max_features=1.0, bootstrap=True, random_state=None):
        super().__init__(n_estimators=n_estimators, max_depth=max_depth,
                         min_samples_leaf=min_samples_leaf,
                         max_features=max_features, bootstrap=bootstrap,
                         random_state=random_state)

    def predict(self, X):
        return self._forest_raw(X).astype(np.float64)


class ExtraTreesRegressor(RandomForestRegressor):
    """Extremely-randomized regression trees (MSE, random thresholds)."""

    _splitter = "random"

    def __init__(self, n_estimators=100, max_depth=10, min_samples_leaf=1,
                 max_features=1.0, bootstrap=False, random_state=None):
        super().__init__(n_estimators=n_estimators, max_depth=max_depth,
                         min_samples_leaf=min_samples_leaf,
                         max_features=max_features, bootstrap=bootstrap,
                         random_state=random_state)


class RFFamily:
    """Engine adapter: per-candidate forests over the shared binned matrix
    (binning shared with the GBT family; trees run back-to-back on device
    with no host round-trips inside a candidate)."""

    name = "rf"

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
        return 32

    @staticmethod
    def make_executor(X, y_enc, classes, device):
        if device is not None:
            from ..ops.rf_executor import DeviceRFExecutor

            return DeviceRFExecutor(X, y_enc, device, classes=classes)
        return _HostRFExecutor(X, y_enc, classes)

    @staticmethod
    def build_plan(estimator, params_list, n_samples, executor, n_classes,
                   error_score):
        from sklearn.base import clone

        proto = clone(estimator)
        base_params = proto.get_params()
        F = executor.F if hasattr(executor, "F") else executor.Xb.shape[1]
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
                specs.append(proto._spec(F=F, n_classes=n_classes))
            except Exception as exc:  # noqa: BLE001 - error_score contract
                if error_score == "raise":
                    raise
                bad[slot] = exc
                specs.append(RFSpec(n_estimators=0, max_depth=1))
        plan = _RFPlan(specs, n_classes)
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
        trees = models[slot]
        est._trees = trees
        est._bin_edges = edges
        if classes is not None:
            est.classes_ = np.asarray(classes)
        est.n_features_in_ = len(edges)
        est.n_iter_ = len(trees)
        return est


class _RFPlan:
    def __init__(self, specs, n_label_classes=0):
        self.specs = specs
        self.col_to_cand = np.arange(len(specs), dtype=np.int32)
        self.n_classes = 1
        self.n_label_classes = n_label_classes

    @property
    def B(self):
        return len(self.specs)


class _HostRFExecutor:
    """Numpy execution with reference semantics (same trees as device)."""

    def __init__(self, X, y_enc, classes=None):
        X = np.asarray(X)
        self.edges = quantile_bins(X)
        self.Xb = bin_data(X, self.edges)
        self.F = self.Xb.shape[1]
        self.y = y_enc
        self.K = len(classes) if classes is not None else 0
        self.msq = 0.0

    supports_sample_weight = True

    def fit(self, train_idx, plan, sample_weight=None):
        Xb_tr = self.Xb[train_idx]
        y_tr = self.y[train_idx]
        sw = (np.asarray(sample_weight, np.float64)[train_idx]
              if sample_weight is not None else None)
        models = []
        n_iter = np.empty(plan.B, np.int32)
        for i, spec in enumerate(plan.specs):
            if spec.n_estimators == 0:
                models.append([])
                n_iter[i] = 0
                continue
            trees = rf_fit_ref(Xb_tr, y_tr, spec, sample_weight=sw)
            models.append(trees)
            n_iter[i] = len(trees)
        return (models, self.edges), n_iter, None

    def scores(self, idx, fitted, plan, metric):
        models, _ = fitted
        Xb = self.Xb[idx]
        yv = self.y[idx]
        out = np.empty(len(models))
        for i, trees in enumerate(models):
            if not trees:
                out[i] = float("nan")
                continue
            if self.K > 2:
                raw = rf_predict_ref(Xb, trees, self.K)
                assert metric == "accuracy"
                out[i] = float(np.mean(raw.argmax(axis=1) ==
                                       yv.astype(np.int64)))
            elif metric == "accuracy":
                raw = rf_predict_ref(Xb, trees)
                out[i] = float(np.mean((raw > 0.5) == (yv > 0.5)))
            elif metric == "r2":
                raw = rf_predict_ref(Xb, trees)
                ss_tot = float(np.sum((yv - yv.mean()) ** 2))
                out[i] = 1.0 - float(np.sum((yv - raw) ** 2)) / \
                    max(ss_tot, 1e-12)
            else:
                raise ValueError(metric)
        return out
