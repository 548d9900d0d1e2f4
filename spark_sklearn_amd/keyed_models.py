"""Per-key many-model training: KeyedEstimator / KeyedModel.

Behavioral counterpart of the reference's ``keyed_models.py`` ~L1-430 [R]
(SURVEY.md C6): group DataFrame rows by ``keyCols``, fit one sklearn
estimator per key, and apply the per-key models row-wise on transform.  The
reference does this over Spark DataFrames with a groupByKey shuffle; here
the frontend is pandas (single node, 288 GB HBM per GPU makes shuffles
moot) and per-key fits route through the same execution machinery as the
search: host fits for arbitrary sklearn estimators, batched CDNA4 kernel
launches for the :mod:`spark_sklearn_amd.models` families (keys batched the
same way candidates are).

>>> import pandas as pd
>>> from sklearn.cluster import KMeans
>>> from spark_sklearn_amd import KeyedEstimator
>>> df = pd.DataFrame({"k": [0, 0, 1, 1],
...                    "features": [[0.0], [0.1], [10.0], [10.1]]})
>>> model = KeyedEstimator(sklearnEstimator=KMeans(n_clusters=1, n_init=2),
...                        keyCols=["k"]).fit(df)
>>> out = model.transform(df)
>>> list(out.columns)
['k', 'features', 'prediction']
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd
import scipy.sparse as sp

from sklearn.base import clone

from .udt import stack_csr_rows

__all__ = ["SparkSklearnEstimator", "KeyedEstimator", "KeyedModel"]


class SparkSklearnEstimator:
    """Thin pickleable wrapper around an sklearn estimator.

    Mirrors the reference's ``SparkSklearnEstimator`` (keyed_models.py [R]):
    attribute access delegates to the wrapped estimator; the wrapper itself
    is what gets stored in the fitted keyed-model table.
    """

    def __init__(self, estimator):
        self._estimator = estimator

    @property
    def estimator(self):
        return self._estimator

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        return getattr(self._estimator, name)

    def __repr__(self):
        return f"SparkSklearnEstimator({self._estimator!r})"


def _stack_feature_column(values) -> np.ndarray:
    """Stack an xCol of per-row vectors (lists, arrays, sparse rows, dicts)
    into a 2-D matrix; sparse rows yield a CSR matrix."""
    vals = list(values)
    if len(vals) == 0:
        raise ValueError("empty group: no rows to stack")
    first = vals[0]
    if sp.issparse(first) or isinstance(first, dict):
        return stack_csr_rows(vals)
    arr = np.asarray(
        [np.asarray(v, dtype=np.float64).ravel() for v in vals])
    return arr


class KeyedEstimator:
    """Fit one estimator per key group of a pandas DataFrame.

    Parameters mirror the reference: ``sklearnEstimator`` (the prototype to
    clone per key), ``keyCols``, ``xCol``, optional ``yCol`` and
    ``estimatorType`` in {"predictor", "transformer", "clusterer"}
    (inferred when omitted: yCol given -> predictor; estimator has
    ``predict`` but no y -> clusterer; else transformer).
    """

    def __init__(self, sklearnEstimator=None, keyCols: Optional[List[str]] = None,
                 xCol: str = "features", yCol: Optional[str] = None,
                 estimatorType: Optional[str] = None):
        if sklearnEstimator is None:
            raise ValueError("sklearnEstimator must be provided")
        if not (hasattr(sklearnEstimator, "fit") and
                hasattr(sklearnEstimator, "get_params")):
            raise ValueError(
                "sklearnEstimator must implement the sklearn estimator "
                "interface (fit/get_params)")
        self.sklearnEstimator = sklearnEstimator
        self.keyCols = list(keyCols) if keyCols else ["key"]
        self.xCol = xCol
        self.yCol = yCol
        if estimatorType is None:
            if yCol is not None:
                estimatorType = "predictor"
            elif hasattr(sklearnEstimator, "predict"):
                estimatorType = "clusterer"
            else:
                estimatorType = "transformer"
        if estimatorType not in ("predictor", "transformer", "clusterer"):
            raise ValueError(f"invalid estimatorType {estimatorType!r}")
        if estimatorType == "predictor" and yCol is None:
            raise ValueError("estimatorType='predictor' requires yCol")
        self.estimatorType = estimatorType

    def fit(self, df: pd.DataFrame) -> "KeyedModel":
        missing = [c for c in self.keyCols + [self.xCol] if c not in df.columns]
        if self.yCol is not None and self.yCol not in df.columns:
            missing.append(self.yCol)
        if missing:
            raise ValueError(f"DataFrame is missing columns {missing}")

        groups = [((key if isinstance(key, tuple) else (key,)), group)
                  for key, group in df.groupby(self.keyCols, sort=True)]
        fitted = self._fit_keys_batched(groups)
        if fitted is None:
            fitted = []
            for key, group in groups:
                X = _stack_feature_column(group[self.xCol].values)
                est = clone(self.sklearnEstimator)
                if self.yCol is not None:
                    est.fit(X, np.asarray(group[self.yCol].values))
                else:
                    est.fit(X)
                fitted.append(key + (SparkSklearnEstimator(est),))
        models = pd.DataFrame.from_records(
            fitted, columns=self.keyCols + ["estimator"])
        return KeyedModel(models, keyCols=self.keyCols, xCol=self.xCol,
                          yCol=self.yCol, estimatorType=self.estimatorType)

    def _fit_keys_batched(self, groups):
        """Batched per-key fits for the linear family on GPU: keys become
        candidate columns of ONE kernel launch, with per-column row masks
        ("per-key fits batched into single kernel launches,
        batch dim = keys").  Per-key sample counts are folded into the
        per-column learning rate and regularization (exact algebra, see
        the scaling below).  Returns None when the fast path does not
        apply (non-linear estimator, no GPU, or key-dependent class sets).
        """
        proto = self.sklearnEstimator
        if getattr(type(proto), "_engine_family", None) != "linear" or \
                self.yCol is None:
            return None
        from .engine import _device_or_none

        device = _device_or_none(None)
        if device is None:
            return None
        from sklearn.base import is_classifier

        ys = [np.asarray(g[self.yCol].values) for _, g in groups]
        if is_classifier(proto):
            classes = np.unique(np.concatenate(ys))
            if any(len(np.unique(y)) != len(classes) for y in ys):
                return None     # keys with missing classes: host loop
            n_classes = len(classes)
        else:
            classes, n_classes = None, 0
        C = n_classes if n_classes > 2 else 1
        if C > 64:
            return None

        Xs = [_stack_feature_column(g[self.xCol].values) for _, g in groups]
        if any(not isinstance(x, np.ndarray) for x in Xs):
            return None     # sparse keyed rows: host loop
        X_all = np.concatenate(Xs).astype(np.float32)
        if classes is not None:
            lut = {v: i for i, v in enumerate(classes)}
            y_all = np.concatenate(
                [np.vectorize(lut.get)(y) for y in ys]).astype(np.int32)
        else:
            y_all = np.concatenate(ys).astype(np.float32)
        n_total = X_all.shape[0]
        row_key = np.repeat(np.arange(len(groups), dtype=np.int32),
                            [len(x) for x in Xs])

        from .models._solver_ref import BatchPlan
        from .ops.device_executor import DeviceLinearExecutor

        executor = DeviceLinearExecutor(X_all, y_all, classes, device)
        fitted = []
        keys_per_chunk = max(1, 256 // C)
        for lo in range(0, len(groups), keys_per_chunk):
            chunk = groups[lo:lo + keys_per_chunk]
            B = len(chunk) * C
            plan = BatchPlan(
                loss=np.empty(B, np.int32), lr=np.empty(B, np.float32),
                l1=np.empty(B, np.float32), l2=np.empty(B, np.float32),
                momentum=np.empty(B, np.float32),
                tol=np.empty(B, np.float32),
                max_iter=np.empty(B, np.int32),
                target_class=np.empty(B, np.int32),
                col_to_cand=np.empty(B, np.int32), n_classes=C,
                fit_intercept=np.empty(B, np.uint8))
            col_key = np.empty(B, np.int32)
            for slot, (key, g) in enumerate(chunk):
                n_k = len(g)
                spec = proto._column_spec(n_k, executor.msq)
                # masked columns see only their key's rows, but the kernel
                # normalizes by n_total: scale lr by n_total/n_k and the
                # penalties by n_k/n_total so the effective update equals
                # the per-key fit exactly
                scale = n_total / max(n_k, 1)
                for c in range(C):
                    j = slot * C + c
                    plan.loss[j] = spec["loss"]
                    plan.lr[j] = spec["lr"] * scale
                    plan.l1[j] = spec["l1"] / scale
                    plan.l2[j] = spec["l2"] / scale
                    plan.momentum[j] = spec["momentum"]
                    plan.tol[j] = spec["tol"]
                    plan.max_iter[j] = spec["max_iter"]
                    plan.fit_intercept[j] = spec["fit_intercept"]
                    plan.target_class[j] = c if C > 1 else \
                        (1 if n_classes else -1)
                    plan.col_to_cand[j] = slot
                    col_key[j] = lo + slot
            (W, b), n_iter, _ = executor.fit(
                np.arange(n_total, dtype=np.int64), plan,
                row_key=row_key, col_key=col_key)
            for slot, (key, g) in enumerate(chunk):
                est = clone(proto)
                cols = slice(slot * C, (slot + 1) * C)
                est._finalize_fit(np.ascontiguousarray(W[:, cols]),
                                  np.ascontiguousarray(b[cols]),
                                  n_iter[cols], classes)
                fitted.append(tuple(key) + (SparkSklearnEstimator(est),))
        return fitted


class KeyedModel:
    """The fitted per-key model table; ``transform`` applies per-row.

    Output column (reference semantics): predictor
    and clusterer append ``prediction``; transformer appends ``output``.
    Rows whose key has no fitted model raise a ``KeyError`` (pass
    ``unseenKeys="drop"`` to silently drop them instead).
    """

    def __init__(self, keyedModels: pd.DataFrame, keyCols: List[str],
                 xCol: str, yCol: Optional[str], estimatorType: str):
        self.keyedModels = keyedModels
        self.keyCols = keyCols
        self.xCol = xCol
        self.yCol = yCol
        self.estimatorType = estimatorType

    @property
    def outputCol(self) -> str:
        return "output" if self.estimatorType == "transformer" else "prediction"

    def transform(self, df: pd.DataFrame,
                  unseenKeys: str = "error") -> pd.DataFrame:
        if unseenKeys not in ("error", "drop"):
            raise ValueError("unseenKeys must be 'error' or 'drop'")
        model_map = {
            tuple(row[c] for c in self.keyCols): row["estimator"].estimator
            for _, row in self.keyedModels.iterrows()
        }
        out_frames = []
        for key, group in df.groupby(self.keyCols, sort=False):
            if not isinstance(key, tuple):
                key = (key,)
            est = model_map.get(key)
            if est is None:
                if unseenKeys == "drop":
                    continue
                raise KeyError(
                    f"no fitted model for key {key!r}; fit() never saw it")
            X = _stack_feature_column(group[self.xCol].values)
            if self.estimatorType == "transformer":
                result = est.transform(X)
                values = [np.asarray(result[i]).ravel()
                          for i in range(len(group))]
            else:
                values = list(np.asarray(est.predict(X)).ravel())
            g = group.copy()
            g[self.outputCol] = values
            out_frames.append(g)
        if not out_frames:
            empty = df.iloc[0:0].copy()
            empty[self.outputCol] = []
            return empty
        return pd.concat(out_frames).sort_index()
