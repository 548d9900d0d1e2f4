import sys
sys.path.insert(0, '/root/repo')
import numpy as np, torch
from spark_sklearn_amd.models._solver_ref import *
from spark_sklearn_amd.models import LogisticRegression
from spark_sklearn_amd.ops.device_executor import DeviceLinearExecutor

rng = np.random.default_rng(9)
Xs, ys = [], []
for k in range(2):
    w = rng.standard_normal(8)
    n_k = 400 + k * 300
    Xk = rng.standard_normal((n_k, 8)).astype(np.float32)
    yk = (Xk @ w > 0).astype(np.int64)
    Xs.append(Xk); ys.append(yk)
X_all = np.concatenate(Xs); y_all = np.concatenate(ys)
n_total = len(X_all)
row_key = np.repeat([0, 1], [len(Xs[0]), len(Xs[1])]).astype(np.int32)
Xb = torch.as_tensor(X_all).to(torch.bfloat16).float().numpy()
msq = float(np.mean(np.sum(Xb.astype(np.float64)**2, axis=1)))

est = LogisticRegression(max_iter=30, tol=0.0)
ex = DeviceLinearExecutor(X_all, y_all.astype(np.float32), np.array([0,1]), torch.device('cuda',0))
print("executor msq", ex.msq, "host msq", msq)

B = 2
mk = lambda v, dt: np.full(B, v, dtype=dt)
specs = [est._column_spec(len(Xs[k]), ex.msq) for k in range(2)]
scale = [n_total/len(Xs[k]) for k in range(2)]
plan = BatchPlan(
    loss=mk(0,np.int32),
    lr=np.array([specs[k]["lr"]*scale[k] for k in range(2)], np.float32),
    l1=mk(0,np.float32),
    l2=np.array([specs[k]["l2"]/scale[k] for k in range(2)], np.float32),
    momentum=mk(0.9,np.float32), tol=mk(0,np.float32),
    max_iter=mk(30,np.int32), target_class=mk(1,np.int32),
    col_to_cand=np.arange(2, dtype=np.int32), n_classes=1, fit_intercept=mk(1,np.uint8))
col_key = np.array([0,1], np.int32)
(W_d, b_d), _, _ = ex.fit(np.arange(n_total, dtype=np.int64), plan,
                          row_key=row_key, col_key=col_key)
# reference per-key fits (bf16 X, executor msq lr)
for k in range(2):
    Xk_b = Xb[row_key == k]
    p1 = BatchPlan(loss=np.array([0],np.int32), lr=np.array([specs[k]["lr"]],np.float32),
                   l1=np.zeros(1,np.float32), l2=np.array([specs[k]["l2"]],np.float32),
                   momentum=np.array([0.9],np.float32), tol=np.zeros(1,np.float32),
                   max_iter=np.array([30],np.int32), target_class=np.array([1],np.int32),
                   col_to_cand=np.zeros(1,np.int32), n_classes=1,
                   fit_intercept=np.ones(1,np.uint8))
    W_r, b_r, _, _ = linear_fit_ref(Xk_b, ys[k], p1)
    print(f"key {k}: dev {W_d[:4, k]} ref {W_r[:4, 0]} match={np.allclose(W_d[:,k], W_r[:,0], rtol=0.05, atol=1e-3)}")
# unmasked full fit comparison (is the mask ignored?)
pf = BatchPlan(loss=np.array([0],np.int32), lr=np.array([auto_lr(ex.msq,0,0)],np.float32),
               l1=np.zeros(1,np.float32), l2=np.zeros(1,np.float32),
               momentum=np.array([0.9],np.float32), tol=np.zeros(1,np.float32),
               max_iter=np.array([30],np.int32), target_class=np.array([1],np.int32),
               col_to_cand=np.zeros(1,np.int32), n_classes=1, fit_intercept=np.ones(1,np.uint8))
W_f, _, _, _ = linear_fit_ref(Xb, y_all, pf)
print("full-fit ref W[:4]", W_f[:4,0])
