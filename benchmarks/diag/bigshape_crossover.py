import json, sys, time
sys.path.insert(0, ".")
import numpy as np, scipy.sparse as sp
import spark_sklearn_amd as ska
from spark_sklearn_amd.models import LogisticRegression
import spark_sklearn_amd.ops.sparse_executor as se

rng = np.random.default_rng(0)
n, f, dens = 10_000_000, 1024, 0.01
Xs = sp.random(n, f, density=dens, format="csr", random_state=1, dtype=np.float32)
w = rng.standard_normal(f).astype(np.float32)
y = (np.asarray(Xs @ w).ravel() + 0.2 * rng.standard_normal(n) > 0).astype(np.int64)

def run(label):
    gs = ska.GridSearchCV(None, LogisticRegression(max_iter=15, tol=0.0),
                          {"C": [0.03, 0.3, 3.0, 30.0]}, cv=3)
    t0 = time.perf_counter(); gs.fit(Xs, y); dt = time.perf_counter() - t0
    print(json.dumps({"path": label, "wall_s": round(dt, 2),
                      "best": round(gs.best_score_, 4)}), flush=True)

se.CSR_DENSITY_LIMIT = 1.0
run("csr")
se.CSR_DENSITY_LIMIT = 0.0
run("dense-ingest")
