import sys
sys.path.insert(0, '/root/repo')
import numpy as np, torch
from spark_sklearn_amd.models._kmeans_ref import KMeansPlan, kmeans_fit_ref, kmeans_init_centers, _assign
from spark_sklearn_amd.ops.kmeans_executor import DeviceKMeansExecutor

rng = np.random.default_rng(3)
centers = rng.standard_normal((5, 48)) * 6
y = rng.integers(0, 5, 20000)
X = (centers[y] + rng.standard_normal((20000, 48))).astype(np.float32)
Xb = torch.as_tensor(X).to(torch.bfloat16).float().numpy()

def mkplan(ks, max_iter):
    cstart = np.cumsum([0] + list(ks)).astype(np.int32)
    nb = len(ks)
    plan = KMeansPlan(cstart=cstart, block_to_cand=np.arange(nb, dtype=np.int32),
                      max_iter=np.full(nb, max_iter, np.int32),
                      tol_scaled=np.full(nb, 1e-4, np.float32), n_cand=nb)
    parts = []
    for i, k in enumerate(ks):
        parts.append(kmeans_init_centers(Xb, k, "k-means++", np.random.default_rng(100+i)))
    plan.C0 = np.concatenate(parts).astype(np.float32)
    return plan

dev = torch.device('cuda', 0)
ex = DeviceKMeansExecutor(X, dev)
train = np.arange(15000, dtype=np.int64)

for iters in (0, 1, 2):
    plan = mkplan([5, 8], iters)
    (C_d, bb, in_d), ni_d, _ = ex.fit(train, plan)
    C_r, ni_r, in_r = kmeans_fit_ref(Xb[train], plan.C0, plan)
    print(f"iters={iters}: dev inertia {in_d}, ref {in_r}, rel {(in_d-in_r)/in_r}")
    print(f"  n_iter dev {list(ni_d)} ref {list(ni_r)}  center maxdiff {np.abs(C_d-C_r).max():.4f}")
