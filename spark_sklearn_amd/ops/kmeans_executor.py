"""Device execution of centroid-block KMeans batches (K2 kernels)."""

from __future__ import annotations

import numpy as np

from . import load_extension
from .device_executor import _as_device_bf16, _roundup, ensure_layout

__all__ = ["DeviceKMeansExecutor"]

INIT_SAMPLE_ROWS = 50_000


class DeviceKMeansExecutor:
    """Executes batched Lloyd fits on one MI355X GPU."""

    def __init__(self, X, device):
        import torch

        self.torch = torch
        self.device = device
        self.ext = load_extension()
        if ensure_layout(device) != 3:
            raise RuntimeError(
                "kmeans kernels are specialized for the MI355X fragment "
                "layout (P=3); probe returned a different layout")
        self.X, self.Xt, self.F = _as_device_bf16(X, device)
        self.Fp = self.X.shape[1]
        self.n = self.X.shape[0]
        self.xnorm = self.ext.row_norms(self.X)
        self.msq = float(self.xnorm.double().mean().item())
        mean_sq_sum = float(
            (self.X.mean(dim=0, dtype=torch.float32) ** 2).sum().item())
        self.feature_mean_var = (self.msq - mean_sq_sum) / self.F
        self._score_state = None

    def init_sample(self, rng) -> np.ndarray:
        """Row sample for host-side seeding, pulled ONCE per executor (the
        per-block rng still drives k-means++ randomness)."""
        if getattr(self, "_init_sample", None) is None:
            import torch

            if self.n <= INIT_SAMPLE_ROWS:
                rows = np.arange(self.n)
            else:
                rows = np.random.default_rng(0).choice(
                    self.n, size=INIT_SAMPLE_ROWS, replace=False)
            rows_t = torch.as_tensor(
                np.asarray(rows, np.int64)).to(self.device)
            self._init_sample = self.X[rows_t, :self.F].float().cpu().numpy()
        return self._init_sample

    def _pad_idx(self, idx):
        import torch

        n = len(idx)
        NTp = _roundup(max(n, 128), 128)
        out = np.zeros(NTp, dtype=np.int32)
        out[:n] = np.asarray(idx, dtype=np.int32)
        return torch.as_tensor(out).to(self.device), n

    supports_sample_weight = True

    def _cached_weights(self, w):
        """Full-length per-row weights, quantized to bf16-representable
        values so the one-hot GEMM numerator and the fp32 weighted-count
        denominator see the SAME w (kernels index by global row id)."""
        hit = getattr(self, "_w_cache", None)
        if hit is not None and hit[0] is w:
            return hit[1]
        torch = self.torch
        t = torch.as_tensor(
            np.ascontiguousarray(np.asarray(w, dtype=np.float32))
        ).to(self.device).to(torch.bfloat16).to(torch.float32)
        self._w_cache = (w, t)
        return t

    def fit(self, train_idx, plan, sample_weight=None):
        torch = self.torch
        idx_t, nt = self._pad_idx(train_idx)
        K_total = int(plan.cstart[-1])
        Kp = min(256, _roundup(max(K_total, 64), 64))
        if K_total > 256:
            raise RuntimeError(
                f"centroid batch {K_total} > 256; reduce the candidate "
                f"chunk (column_batch)")
        nb = plan.n_blocks

        C0 = np.zeros((Kp, self.Fp), dtype=np.float32)
        C0[:K_total, :self.F] = plan.C0
        cen_block = np.full(Kp, nb, dtype=np.int32)
        for b in range(nb):
            cen_block[plan.cstart[b]:plan.cstart[b + 1]] = b

        dev = self.device
        C0_t = torch.as_tensor(C0).to(dev)
        cstart_t = torch.as_tensor(
            np.asarray(plan.cstart, np.int32)).to(dev)
        cen_block_t = torch.as_tensor(cen_block).to(dev)
        mi_t = torch.as_tensor(
            np.asarray(plan.max_iter, np.int32)).to(dev)
        tol_t = torch.as_tensor(
            np.asarray(plan.tol_scaled, np.float32)).to(dev)

        max_epochs = int(plan.max_iter.max()) if nb else 0
        check_every = 8 if max_epochs > 12 else 0
        sw_t = (self._cached_weights(sample_weight)
                if sample_weight is not None else
                torch.empty(0, dtype=torch.float32, device=dev))
        C, n_iter, inertia, Ct, cnorm, labels = self.ext.kmeans_fit(
            self.X, self.Xt, idx_t, C0_t, cstart_t, cen_block_t, mi_t,
            tol_t, self.xnorm, nt, max_epochs, check_every, sw_t)
        torch.cuda.synchronize(dev)
        self._score_state = (Ct, cnorm, cstart_t)

        inertia_np = inertia.double().cpu().numpy()
        best_block = np.empty(plan.n_cand, np.int32)
        for c in range(plan.n_cand):
            blocks = np.flatnonzero(plan.block_to_cand == c)
            best_block[c] = blocks[np.argmin(inertia_np[blocks])]
        C_np = C[:K_total, :self.F].cpu().numpy()
        return ((C_np, best_block, inertia_np),
                n_iter.cpu().numpy(), inertia_np)

    def scores(self, idx, fitted, plan, metric):
        assert metric == "neg_inertia"
        if self._score_state is None:
            raise RuntimeError("scores() requires a preceding fit()")
        Ct, cnorm, cstart_t = self._score_state
        _, best_block, _ = fitted
        idx_t, nt = self._pad_idx(idx)
        inertia, _ = self.ext.kmeans_assign(
            self.X, idx_t, Ct, cnorm, self.xnorm, cstart_t, nt, 0)
        self.torch.cuda.synchronize(self.device)
        inertia_np = inertia.double().cpu().numpy()
        return np.array([-inertia_np[int(best_block[c])]
                         for c in range(plan.n_cand)])

    def assign_labels(self, idx, centers) -> np.ndarray:
        torch = self.torch
        centers = np.asarray(centers, dtype=np.float32)
        k = centers.shape[0]
        Kp = min(256, _roundup(max(k, 64), 64))
        C = np.zeros((Kp, self.Fp), dtype=np.float32)
        C[:k, :self.F] = centers
        Ct = torch.as_tensor(C).to(self.device).to(torch.bfloat16)
        from ..models._kmeans_ref import bf16_round_arr

        cb = bf16_round_arr(centers.astype(np.float32)).astype(np.float64)
        cnorm = torch.as_tensor(
            np.concatenate([np.sum(cb ** 2, axis=1),
                            np.full(Kp - k, 3e38)]).astype(np.float32)
        ).to(self.device)
        cstart = torch.as_tensor(
            np.array([0, k], np.int32)).to(self.device)
        idx_t, nt = self._pad_idx(idx)
        _, labels = self.ext.kmeans_assign(
            self.X, idx_t, Ct, cnorm, self.xnorm, cstart, nt, 1)
        torch.cuda.synchronize(self.device)
        return labels[0, :nt].cpu().numpy().astype(np.int32)
