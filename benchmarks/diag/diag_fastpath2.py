import sys
sys.path.insert(0, '/root/repo')
import numpy as np, torch
import spark_sklearn_amd.engine as eng
from spark_sklearn_amd.models import LogisticRegression
from spark_sklearn_amd.models._solver_ref import BatchPlan
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
plan, _ = eng.LinearFamily.build_plan(est, [{"C": 0.01}, {"C": 0.1}, {"C": 1.0}, {"C": 10.0}],
                                      len(splits[0][0]), ex, 2, "raise")
Bt = 256
def pad(src, fill, dt):
    out = np.full(Bt, fill, dtype=dt); out[:len(src)] = src
    return torch.as_tensor(out).to(ex.device)
loss_t = pad(plan.loss, 0, np.int32); lr_t = pad(plan.lr, 0, np.float32)
l1_t = pad(plan.l1, 0, np.float32); l2_t = pad(plan.l2, 0, np.float32)
mom_t = pad(plan.momentum, 0, np.float32); tol_t = pad(plan.tol, 0, np.float32)
mi_t = pad(plan.max_iter, 0, np.int32); tgt_t = pad(plan.target_class, 0, np.int32)
ic_t = pad(plan.fit_intercept, 0, np.uint8)

for fold_sel in ([0], [1], [2], [0,1,2]):
    train_ts, nts, test_ts = [], [], []
    for fi in fold_sel:
        tr, te = splits[fi]
        train_ts.append(ex._cached_idx(tr, 256, np.int32)); nts.append(len(tr))
        test_ts.append(ex._cached_idx(te, 0, np.int64))
    out = ex.ext.linear_search_run(ex.X, ex.Xt, ex.y, train_ts, nts, test_ts,
        loss_t, lr_t, l1_t, l2_t, mom_t, tol_t, mi_t, tgt_t, ic_t,
        1, _METRIC_ID["accuracy"], 4, 0, 32)
    ts, _, fitms, _, ni = out
    acc = (ts.cpu().numpy() / np.array([len(splits[fi][1]) for fi in fold_sel])[:, None])
    print(f"folds {fold_sel}: acc {np.round(acc, 3).tolist()} n_iter {ni.cpu().numpy()[:, :4].tolist()}")
