"""Device execution of RandomForest candidate chunks (K3b kernels).

Shares the resident uint8 binned matrix (and its cache) with the GBT
executor; per candidate, `rf_fit` runs the whole forest on device --
bootstrap resample, per-level histograms with sibling subtraction,
gini/MSE split scan with exact-k feature subsampling, leaf
probabilities -- with no host round-trips between trees.
"""

from __future__ import annotations

import numpy as np

from .gbt_executor import DeviceGBTExecutor

__all__ = ["DeviceRFExecutor"]


class _LazyForests:
    """List-like view over the device forests: the numpy tree tuples that
    install_fitted / host scoring need are materialized per slot ON
    DEMAND.  The search scores everything on device, so eagerly pulling
    [T*K][NN] planes to host and building up to T*K python tuples per
    candidate (the old fit path) was pure overhead -- it dominated the
    wall clock of the launch-bound digits-grid workload."""

    def __init__(self, models_dev, K):
        self._dev = models_dev
        self._K = K
        self._cache = {}

    def __len__(self):
        return len(self._dev)

    def __getitem__(self, i):
        if i in self._cache:
            return self._cache[i]
        m = self._dev[i]
        if m is None:
            trees = []
        else:
            _, feats, thrs, vals, spec = m
            K, T = self._K, spec.n_estimators
            f_np = feats.cpu().numpy()
            t_np = thrs.cpu().numpy().astype(np.uint8)
            v_np = vals.cpu().numpy()
            if K > 2:
                trees = [[(f_np[t * K + c], t_np[t * K + c],
                           v_np[t * K + c]) for c in range(K)]
                         for t in range(T)]
            else:
                trees = [(f_np[t], t_np[t], v_np[t]) for t in range(T)]
        self._cache[i] = trees
        return trees


class DeviceRFExecutor(DeviceGBTExecutor):
    supports_sample_weight = True

    def fit(self, train_idx, plan, sample_weight=None):
        torch = self.torch
        idx_np = np.asarray(train_idx, dtype=np.int32)
        idx_t = torch.as_tensor(idx_np).to(self.device)
        y_tr_np = self.y_np[idx_np]
        y_tr = torch.as_tensor(y_tr_np).to(self.device)
        sw_t = torch.empty(0, dtype=torch.float32, device=self.device)
        if sample_weight is not None:
            sw_t = torch.as_tensor(
                np.asarray(sample_weight, np.float32)[idx_np]
            ).to(self.device)

        models_dev = []
        n_iter = np.empty(plan.B, np.int32)
        K = self.K
        for i, spec in enumerate(plan.specs):
            if spec.n_estimators == 0:
                models_dev.append(None)
                n_iter[i] = 0
                continue
            feats, thrs, vals = self.ext.rf_fit(
                self.Xb, self.Xbt, idx_t, y_tr, K, spec.n_estimators,
                spec.max_depth,
                spec.min_samples_leaf, spec.max_features,
                spec.random_state, 1 if spec.bootstrap else 0,
                1 if spec.splitter == "random" else 0, sw_t)
            kind = "multi" if K > 2 else "flat"
            models_dev.append((kind, feats, thrs, vals, spec))
            n_iter[i] = spec.n_estimators
        torch.cuda.synchronize(self.device)
        self._models_dev = models_dev
        return (_LazyForests(models_dev, K), self.edges), n_iter, None

    def scores(self, idx, fitted, plan, metric):
        torch = self.torch
        if self._models_dev is None:
            raise RuntimeError("scores() requires a preceding fit()")
        idx_t = torch.as_tensor(
            np.asarray(idx, dtype=np.int32)).to(self.device)
        n = len(idx)
        out = np.empty(len(self._models_dev))
        mid = 0 if metric == "accuracy" else 1
        if metric == "r2":
            y_sel = self.y_np[np.asarray(idx)]
            ss_tot = float(np.sum((y_sel - y_sel.mean()) ** 2))
        # queue every candidate's metric on device, then ONE sync + D2H
        # at the end (per-candidate .item() was a sync per candidate)
        sums_dev = [None] * len(self._models_dev)
        for i, m in enumerate(self._models_dev):
            if m is None:
                continue
            kind, feats, thrs, vals, spec = m
            if kind == "multi":
                if getattr(self, "_zeros_k", None) is None:
                    self._zeros_k = torch.zeros(
                        self.K, dtype=torch.float32, device=self.device)
                raws = self.ext.gbt_predict_multi(
                    self.Xb, idx_t, feats, thrs, vals, self.K,
                    self._zeros_k)
                sums_dev[i] = self.ext.gbt_metric_multi(raws, self.y,
                                                        idx_t, self.K)
                continue
            # binary: leaf vals are p/T; raw + f0(-0.5) > 0 <=> mean p > .5
            f0 = -0.5 if metric == "accuracy" else 0.0
            raw = self.ext.gbt_predict(self.Xb, idx_t, feats, thrs, vals,
                                       f0)
            sums_dev[i] = self.ext.gbt_metric(raw, self.y, idx_t, mid)
        live = [s for s in sums_dev if s is not None]
        vs = torch.cat(live).cpu().numpy() if live else np.empty(0)
        j = 0
        for i, s in enumerate(sums_dev):
            if s is None:
                out[i] = float("nan")
                continue
            v = float(vs[j])
            j += 1
            if self.K > 2 or metric == "accuracy":
                out[i] = v / n
            else:
                out[i] = 1.0 - v / max(ss_tot, 1e-12)
        return out
