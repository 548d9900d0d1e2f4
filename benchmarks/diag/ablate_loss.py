import sys, time
sys.path.insert(0, '/root/repo')
import numpy as np, torch
from spark_sklearn_amd.models._solver_ref import BatchPlan, LOSS_LOGISTIC, LOSS_HINGE, LOSS_SQUARED
from spark_sklearn_amd.ops.device_executor import DeviceLinearExecutor

rng = np.random.default_rng(0)
n, f = 1_000_000, 256
X = rng.standard_normal((n, f), dtype=np.float32)
y = (X[:, 0] > 0).astype(np.float32)
dev = torch.device('cuda', 0)
ex = DeviceLinearExecutor(X, y, None, dev)
train = np.arange(800_000, dtype=np.int64)

def mkplan(loss, B=256, iters=25):
    mk = lambda v, dt: np.full(B, v, dtype=dt)
    return BatchPlan(loss=mk(loss, np.int32), lr=mk(1e-3, np.float32),
                     l1=mk(0, np.float32), l2=mk(1e-6, np.float32),
                     momentum=mk(0.9, np.float32), tol=mk(0, np.float32),
                     max_iter=mk(iters, np.int32), target_class=mk(1, np.int32),
                     col_to_cand=np.arange(B, dtype=np.int32), n_classes=1,
                     fit_intercept=mk(1, np.uint8))

for name, lid in [('logistic', LOSS_LOGISTIC), ('hinge', LOSS_HINGE), ('squared', LOSS_SQUARED)]:
    plan = mkplan(lid)
    ex.fit(train, plan)  # warm
    t0 = time.perf_counter()
    for _ in range(3):
        ex.fit(train, plan)
    dt = (time.perf_counter() - t0) / 3
    print(f"{name}: {dt*1000:.1f} ms per 25-epoch 256-col fit  ({dt/25*1000:.2f} ms/epoch)")
