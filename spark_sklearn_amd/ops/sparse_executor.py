"""CSR-resident device execution for the linear family (K5 kernels).

Used when the input is scipy-sparse, low-density AND too large for dense
residency (the capacity gate in engine._new_device_executor; see
profiles/r02_csr_crossover.md -- dense-ingest measured faster wherever
the dense form fits).  The matrix stays in HBM as CSR
(indptr/indices/data) plus a (row-chunk, feature, row)-sorted transpose
entry list for the backward pass -- never densified.  Targets: binary,
regression, and multiclass-OVR (accuracy / neg_log_loss via the
per-candidate argmax kernel csr_scores_multi_kernel).
"""

from __future__ import annotations

import numpy as np
import scipy.sparse as sp

from . import load_extension
from .device_executor import _METRIC_ID, _roundup, ensure_layout

__all__ = ["SparseLinearExecutor", "CSR_DENSITY_LIMIT"]

CSR_DENSITY_LIMIT = 0.05      # CSR candidate at or below this density
# Round-2 measurement (profiles/r02_csr_crossover.md): the dense-ingest
# path beat the CSR-resident kernels at EVERY tested (shape, density) --
# 2M x 512 at 0.5-10% and the headline 10M x 1024 at 1% (2.16 s vs
# 4.99 s) -- because the batched bf16 MFMA kernels amortize the dense
# matrix over 256 candidate columns.  The CSR path's remaining role is
# CAPACITY: matrices whose dense bf16 form (plus transpose) would not
# fit comfortably in 288 GB HBM.
CSR_MIN_DENSE_BYTES = 64e9    # use CSR only above this dense footprint
BWD_CHUNK_ROWS = 4096         # D rows touched by a chunk stay L2-resident

_sparse_cache: dict = {}


class SparseLinearExecutor:
    """Executes batched linear fits/scores over a resident CSR matrix."""

    def __init__(self, X, y_enc, classes, device):
        import torch

        self.torch = torch
        self.device = device
        self.ext = load_extension()
        if ensure_layout(device) != 3:
            raise RuntimeError("CSR kernels require the probed layout P=3")
        self.C = len(classes) if (classes is not None and
                                  len(classes) > 2) else 1

        key = (id(X), X.shape, str(device))
        hit = _sparse_cache.get(key)
        if hit is None or hit[0] is not X:
            Xc = sp.csr_matrix(X, dtype=np.float32)
            Xc.sort_indices()
            n, F = Xc.shape
            dev = device
            indptr = torch.as_tensor(
                Xc.indptr.astype(np.int32)).to(dev)
            cols = torch.as_tensor(
                Xc.indices.astype(np.int32)).to(dev)
            vals = torch.as_tensor(Xc.data).to(dev).to(torch.bfloat16)
            # bwd structure: entries sorted by (row chunk, feature, row).
            # The sort runs ON DEVICE: a host np.lexsort over 100M nnz
            # costs ~10 s; torch.argsort on a unique composite int64 key
            # costs ~50 ms.  key = ((chunk*F + col)*n + row) is unique per
            # entry and fits int64 far beyond the 288 GB HBM shapes.
            coo = Xc.tocoo()
            nchunks = (n - 1) // BWD_CHUNK_ROWS + 1 if n else 1
            col64 = coo.col.astype(np.int64)
            row64 = coo.row.astype(np.int64)
            skey = ((row64 // BWD_CHUNK_ROWS) * F + col64) * n + row64
            key_t = torch.as_tensor(skey).to(dev)
            order_t = torch.argsort(key_t)
            ks = key_t[order_t]
            tf = torch.as_tensor(
                coo.col.astype(np.int32)).to(dev)[order_t].contiguous()
            tr = torch.as_tensor(
                coo.row.astype(np.int32)).to(dev)[order_t].contiguous()
            tv = torch.as_tensor(coo.data).to(dev)[order_t] \
                .to(torch.bfloat16).contiguous()
            chunk_sorted = ks // (np.int64(F) * np.int64(n))
            chunk_off = torch.searchsorted(
                chunk_sorted,
                torch.arange(nchunks + 1, dtype=torch.int64,
                             device=dev)).to(torch.int64).contiguous()
            del key_t, order_t, ks, chunk_sorted
            sq = np.asarray(Xc.multiply(Xc).sum(axis=1)).ravel()
            msq = float(np.mean(sq))
            if len(_sparse_cache) > 2:
                _sparse_cache.clear()
            _sparse_cache[key] = (X, (indptr, cols, vals, tf, tr, tv,
                                      chunk_off, n, F, msq))
            hit = _sparse_cache[key]
        (self.indptr, self.cols, self.vals, self.tf, self.tr, self.tv,
         self.chunk_off, self.n, self.F, self.msq) = hit[1]
        self.y = torch.as_tensor(
            np.asarray(y_enc, dtype=np.float32)).to(device)
        self._fit_state = None
        self._inv_cache: dict = {}
        self._idx_cache: dict = {}

    def _cached(self, cache, arr, builder):
        key = (id(arr), len(arr))
        hit = cache.get(key)
        if hit is not None and hit[0] is arr:
            return hit[1]
        t = builder(arr)
        if len(cache) > 32:
            cache.clear()
        cache[key] = (arr, t)
        return t

    def fit(self, train_idx, plan):
        torch = self.torch
        nt = len(train_idx)
        B = plan.B
        # csr_linear_fit contract: Bp in {64, 128, 256}
        Bp = 64 if B <= 64 else (128 if B <= 128 else 256)
        if B > 256:
            raise RuntimeError("chunk larger batches to <= 256 columns")

        def pad_i32(a, fill=0):
            out = np.full(Bp, fill, dtype=np.int32)
            out[:B] = np.asarray(a, dtype=np.int32)
            return torch.as_tensor(out).to(self.device)

        def pad_f32(a, fill=0.0):
            out = np.full(Bp, fill, dtype=np.float32)
            out[:B] = np.asarray(a, dtype=np.float32)
            return torch.as_tensor(out).to(self.device)

        idx_t = self._cached(
            self._idx_cache, train_idx,
            lambda a: torch.as_tensor(
                np.asarray(a, dtype=np.int32)).to(self.device))

        def build_inv(a):
            inv = np.full(self.n, -1, dtype=np.int32)
            inv[np.asarray(a, dtype=np.int64)] = np.arange(
                len(a), dtype=np.int32)
            return torch.as_tensor(inv).to(self.device)

        inv_t = self._cached(self._inv_cache, train_idx, build_inv)

        ic_np = np.zeros(Bp, dtype=np.uint8)
        ic_np[:B] = np.asarray(plan.fit_intercept, dtype=np.uint8)
        Fp = max(64, _roundup(self.F, 64))
        max_epochs = int(plan.max_iter.max()) if B else 0
        W, bias, n_iter, cur_loss, Wf = self.ext.csr_linear_fit(
            self.indptr, self.cols, self.vals, self.tf, self.tr, self.tv,
            self.chunk_off, inv_t, self.y, idx_t,
            pad_i32(plan.loss), pad_f32(plan.lr), pad_f32(plan.l1),
            pad_f32(plan.l2), pad_f32(plan.momentum), pad_f32(plan.tol),
            pad_i32(plan.max_iter), pad_i32(plan.target_class),
            torch.as_tensor(ic_np).to(self.device),
            Fp, max_epochs, 32 if max_epochs > 48 else 0)
        torch.cuda.synchronize(self.device)
        self._fit_state = (Wf, bias, pad_i32(plan.loss),
                           pad_i32(plan.target_class))
        W_np = W[:B, :self.F].float().cpu().numpy().T.copy()
        return ((W_np, bias[:B].float().cpu().numpy()),
                n_iter[:B].cpu().numpy(),
                cur_loss[:B].float().cpu().numpy())

    def scores(self, idx, fitted, plan, metric):
        torch = self.torch
        if self._fit_state is None:
            raise RuntimeError("scores() requires a preceding fit()")
        Wf, bias, loss_t, tgt_t = self._fit_state
        n_cand = int(plan.col_to_cand.max()) + 1
        C = plan.n_classes
        if C != 1 and metric == "r2":
            raise RuntimeError("r2 scoring is single-output only")
        idx_t = torch.as_tensor(
            np.asarray(idx, dtype=np.int64)).to(self.device)
        sums = self.ext.csr_linear_scores(
            self.indptr, self.cols, self.vals, self.y, idx_t, Wf, bias,
            loss_t, tgt_t, _METRIC_ID[metric], n_cand, C).cpu().numpy()
        n = len(idx)
        if metric in ("accuracy", "neg_log_loss"):
            return sums / n
        y_sel = self.y[idx_t].cpu().numpy()
        ss_tot = float(np.sum((y_sel - y_sel.mean()) ** 2))
        if ss_tot <= 0:
            return np.zeros(n_cand)
        return 1.0 - sums / ss_tot
