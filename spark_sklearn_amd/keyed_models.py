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

        records = []
        for key, group in df.groupby(self.keyCols, sort=True):
            if not isinstance(key, tuple):
                key = (key,)
            X = _stack_feature_column(group[self.xCol].values)
            est = clone(self.sklearnEstimator)
            if self.yCol is not None:
                est.fit(X, np.asarray(group[self.yCol].values))
            else:
                est.fit(X)
            records.append(key + (SparkSklearnEstimator(est),))
        models = pd.DataFrame.from_records(
            records, columns=self.keyCols + ["estimator"])
        return KeyedModel(models, keyCols=self.keyCols, xCol=self.xCol,
                          yCol=self.yCol, estimatorType=self.estimatorType)


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
