import sys
sys.path.insert(0, '/root/repo')
import numpy as np, torch
from spark_sklearn_amd.models._solver_ref import *
from spark_sklearn_amd.ops.device_executor import DeviceLinearExecutor

rng = np.random.default_rng(3)
n, f = 4096, 96   # Fp=128 -> bwd_g5;  also test f=256 -> bwd_g8
for f in (96, 256):
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
    y = ((X @ w) > 0).astype(np.int64)
    Xb = torch.as_tensor(X).to(torch.bfloat16).float().numpy()
    msq = float(np.mean(np.sum(Xb.astype(np.float64)**2, axis=1)))
    lr = auto_lr(msq, LOSS_LOGISTIC, 0.0)
    B = 256
    mk = lambda v, dt: np.full(B, v, dtype=dt)
    plan = BatchPlan(loss=mk(0, np.int32), lr=mk(lr, np.float32),
                     l1=mk(0, np.float32),
                     l2=np.logspace(-6, -2, B).astype(np.float32),
                     momentum=mk(0.9, np.float32), tol=mk(0, np.float32),
                     max_iter=mk(6, np.int32), target_class=mk(1, np.int32),
                     col_to_cand=np.arange(B, dtype=np.int32), n_classes=1,
                     fit_intercept=mk(1, np.uint8))
    ex = DeviceLinearExecutor(X, y.astype(np.float32), None, torch.device('cuda', 0))
    train = np.arange(0, 3500, dtype=np.int64)
    (W_d, b_d), ni, _ = ex.fit(train, plan)
    W_r, b_r, _, _ = linear_fit_ref(Xb[train], y[train], plan)
    err = np.abs(W_d - W_r).max()
    rel = err / (np.abs(W_r).max() + 1e-12)
    print(f"F={f}: Bp=256 fit |dW|max={err:.5f} rel={rel:.4f} "
          f"|W_ref|max={np.abs(W_r).max():.4f} bias diff={np.abs(b_d-b_r).max():.6f}")
    # per-column profile of error (which cols wrong?)
    colerr = np.abs(W_d - W_r).max(axis=0)
    bad = np.flatnonzero(colerr > 0.05 * np.abs(W_r).max())
    print(f"   bad cols ({len(bad)}):", bad[:16], "..." if len(bad) > 16 else "")
