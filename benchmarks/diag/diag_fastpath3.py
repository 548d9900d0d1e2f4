import sys
sys.path.insert(0, '/root/repo')
import numpy as np, torch
import spark_sklearn_amd.engine as eng
from spark_sklearn_amd.models import LogisticRegression
from spark_sklearn_amd.ops.device_executor import DeviceLinearExecutor, _METRIC_ID
from sklearn.model_selection import KFold

rng = np.random.default_rng(7)
n, f = 8192, 96
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)
classes, y_enc = eng.LinearFamily.encode_targets(LogisticRegression(), y)
ex = DeviceLinearExecutor(X, y_enc, classes, torch.device('cuda', 0))
splits = list(KFold(3).split(X))
est = LogisticRegression(max_iter=15, tol=0.0)
params = [{"C": 0.01}, {"C": 0.1}, {"C": 1.0}, {"C": 10.0}]
plan, _ = eng.LinearFamily.build_plan(est, params, len(splits[0][0]), ex, 2, "raise")

# proven path
(W_ref, b_ref), ni, _ = ex.fit(splits[0][0], plan)
print("linear_fit: |W| =", np.abs(W_ref).max(), "b:", b_ref[:4])

Bt = 256
def pad(src, fill, dt):
    out = np.full(Bt, fill, dtype=dt); out[:len(src)] = src
    return torch.as_tensor(out).to(ex.device)
args = [pad(plan.loss,0,np.int32), pad(plan.lr,0,np.float32), pad(plan.l1,0,np.float32),
        pad(plan.l2,0,np.float32), pad(plan.momentum,0,np.float32), pad(plan.tol,0,np.float32),
        pad(plan.max_iter,0,np.int32), pad(plan.target_class,0,np.int32), pad(plan.fit_intercept,0,np.uint8)]
tr_t = ex._cached_idx(splits[0][0], 256, np.int32)
te_t = ex._cached_idx(splits[0][1], 0, np.int64)
out = ex.ext.linear_search_run(ex.X, ex.Xt, ex.y, [tr_t], [len(splits[0][0])], [te_t],
                               *args, 1, _METRIC_ID["accuracy"], 4, 0, 32)
W_run = out[5][:4, :ex.F].cpu().numpy().T
b_run = out[6][:4].cpu().numpy()
print("search_run: |W| =", np.abs(W_run).max(), "b:", b_run)
print("W allclose:", np.allclose(W_run, W_ref, rtol=1e-3, atol=1e-5))
print("acc sums:", (out[0].cpu().numpy() / len(splits[0][1])).round(4))
