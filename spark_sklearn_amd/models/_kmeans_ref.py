"""Reference batched KMeans (Lloyd) solver -- numpy, fp32.

Defines the exact semantics of the K2 HIP kernels:
candidates batch as CENTROID BLOCKS -- all candidates' (and all n_init
restarts') centroids concatenate into one C_all [K_total, F] matrix, so the
per-iteration distance computation for every candidate is ONE MFMA GEMM
S = X . C_all^T, followed by per-candidate argmin over its block slice and
block-sliced centroid accumulation.  This is the same batching move as the
linear family's candidate columns.

Semantics follow sklearn KMeans (lloyd):
  * distances argmin per row -> labels; ties to the lower index
  * new center = mean of assigned rows; an EMPTY cluster keeps its old
    center (divergence from sklearn, which relocates empty clusters to
    far points; documented -- empty clusters are rare at sane k)
  * convergence: squared Frobenius norm of the center shift <= tol_scaled,
    where tol_scaled = tol * mean(var(X, axis=0)) (sklearn's scaling)
  * inertia = sum of squared distances to the assigned center
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

__all__ = ["KMeansPlan", "kmeans_init_centers", "kmeans_fit_ref",
           "kmeans_score_ref"]


@dataclass
class KMeansPlan:
    """Centroid-block layout for one candidate batch.

    ``cstart`` has length n_blocks+1; block b owns centroid rows
    [cstart[b], cstart[b+1]) of C_all.  A candidate with n_init restarts
    owns ``n_init`` consecutive blocks (``block_to_cand`` maps back);
    the best restart by train inertia represents the candidate.
    """

    cstart: np.ndarray        # int32 [n_blocks + 1]
    block_to_cand: np.ndarray  # int32 [n_blocks]
    max_iter: np.ndarray      # int32 [n_blocks]
    tol_scaled: np.ndarray    # float32 [n_blocks]
    n_cand: int

    @property
    def n_blocks(self) -> int:
        return len(self.block_to_cand)

    @property
    def K_total(self) -> int:
        return int(self.cstart[-1])


def kmeans_init_centers(X_sample: np.ndarray, k: int, init: str,
                        rng: np.random.Generator) -> np.ndarray:
    """Host-side center init on a row sample (k-means++ or random).

    k-means++ runs on a sample (<=50k rows by the callers) -- the classic
    seeding quality is preserved while keeping init off the hot path.
    """
    n = X_sample.shape[0]
    if k > n:
        raise ValueError(f"n_clusters={k} > usable rows {n}")
    if init == "random":
        rows = rng.choice(n, size=k, replace=False)
        return X_sample[rows].astype(np.float32)
    if init != "k-means++":
        raise ValueError(f"unsupported init {init!r}")
    centers = np.empty((k, X_sample.shape[1]), dtype=np.float32)
    centers[0] = X_sample[rng.integers(n)]
    closest = np.full(n, np.inf, dtype=np.float64)
    # ||x-c||^2 via the GEMV expansion: one BLAS matvec per round instead
    # of a broadcast temp the size of the sample (6x faster seeding)
    xsq = np.einsum("ij,ij->i", X_sample, X_sample)
    for i in range(1, k):
        c = centers[i - 1]
        d = xsq - 2.0 * (X_sample @ c) + np.dot(c, c)
        np.maximum(d, 0.0, out=d)
        np.minimum(closest, d, out=closest)
        total = closest.sum()
        if total <= 0:
            centers[i] = X_sample[rng.integers(n)]
            continue
        r = rng.random() * total
        centers[i] = X_sample[np.searchsorted(np.cumsum(closest), r)]
    return centers


def bf16_round_arr(a: np.ndarray) -> np.ndarray:
    """Round fp32 to bf16-representable values (round-to-nearest-even).

    Distances are DEFINED against bf16-rounded centers in both tiers: the
    device GEMM consumes the bf16 centroid mirror, so the reference rounds
    identically -- keeps the two bit-consistent instead of hiding a
    systematic 0.2%-of-|S| skew inside test tolerances."""
    u = np.ascontiguousarray(a, dtype=np.float32).view(np.uint32)
    r = ((u + 0x7FFF + ((u >> 16) & 1)) >> 16) << 16
    return r.astype(np.uint32).view(np.float32)


def _assign(X, C_all, plan, w=None):
    """labels [n, n_blocks] int32 + per-block inertia (float64);
    w weights the inertia terms (assignment itself is weight-free)."""
    C_all = bf16_round_arr(C_all)
    S = X @ C_all.T                                   # [n, K_total]
    cn = np.sum(C_all.astype(np.float64) ** 2, axis=1)
    xn = np.sum(X.astype(np.float64) ** 2, axis=1)
    n = X.shape[0]
    labels = np.empty((n, plan.n_blocks), dtype=np.int32)
    inertia = np.zeros(plan.n_blocks, dtype=np.float64)
    for b in range(plan.n_blocks):
        lo, hi = int(plan.cstart[b]), int(plan.cstart[b + 1])
        q = cn[lo:hi][np.newaxis, :] - 2.0 * S[:, lo:hi]
        lab = np.argmin(q, axis=1)
        labels[:, b] = lab
        terms = xn + q[np.arange(n), lab]
        if w is not None:
            terms = terms * w
        inertia[b] = float(np.sum(terms))
    return labels, inertia


def kmeans_fit_ref(X: np.ndarray, C0_all: np.ndarray, plan: KMeansPlan,
                   sample_weight=None):
    """Run Lloyd iterations for all blocks simultaneously.

    Returns (C_all [K_total, F] fp32, n_iter [n_blocks], inertia
    [n_blocks] -- inertia of the FINAL assignment).

    sample_weight: weighted centroid means / inertia (assignment is
    weight-free).  The device path quantizes w to bf16-representable
    values for its one-hot GEMM; pass the same through bf16_round_arr
    when checking parity.
    """
    X = np.ascontiguousarray(X, dtype=np.float32)
    C_all = np.array(C0_all, dtype=np.float32, copy=True)
    w = None if sample_weight is None else \
        np.asarray(sample_weight, dtype=np.float64)
    nb = plan.n_blocks
    active = np.ones(nb, dtype=bool)
    n_iter = np.zeros(nb, dtype=np.int32)
    inertia = np.zeros(nb, dtype=np.float64)

    max_epochs = int(plan.max_iter.max()) if nb else 0
    for it in range(max_epochs):
        epoch_active = active & (it < plan.max_iter)
        if not epoch_active.any():
            break
        labels, cur_inertia = _assign(X, C_all, plan, w)
        for b in range(nb):
            if not epoch_active[b]:
                continue
            lo, hi = int(plan.cstart[b]), int(plan.cstart[b + 1])
            k = hi - lo
            sums = np.zeros((k, X.shape[1]), dtype=np.float64)
            if w is None:
                counts = np.bincount(labels[:, b],
                                     minlength=k).astype(np.float64)
                np.add.at(sums, labels[:, b], X)
            else:
                counts = np.bincount(labels[:, b], weights=w,
                                     minlength=k).astype(np.float64)
                np.add.at(sums, labels[:, b], X * w[:, np.newaxis])
            new = C_all[lo:hi].astype(np.float64)
            nz = counts > 0
            new[nz] = sums[nz] / counts[nz, np.newaxis]
            shift = float(np.sum((new - C_all[lo:hi]) ** 2))
            C_all[lo:hi] = new.astype(np.float32)
            n_iter[b] = it + 1
            inertia[b] = cur_inertia[b]
            if shift <= plan.tol_scaled[b]:
                active[b] = False
    # final inertia under the FINAL centers
    _, inertia = _assign(X, C_all, plan, w)
    return C_all, n_iter, inertia


def kmeans_score_ref(X: np.ndarray, C_all: np.ndarray,
                     plan: KMeansPlan, best_block: np.ndarray) -> np.ndarray:
    """Per-candidate score = -inertia(X) of the candidate's best block."""
    X = np.ascontiguousarray(X, dtype=np.float32)
    _, inertia = _assign(X, C_all, plan)
    return np.array([-inertia[int(best_block[c])]
                     for c in range(plan.n_cand)])
