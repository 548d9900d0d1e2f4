"""Multi-process CPU tier: gloo backend, world_size=2 (BASELINE config 1).

Runs the full distributed search path -- RC1 broadcast, candidate-partition,
local fits, RC2 score all-reduce, rank-0 refit + model broadcast -- with no
GPU, proving the distributed path is correct by construction before it ever
touches RCCL (the GPU path runs the identical code with backend "nccl").
"""

import json
import os
import subprocess
import sys
import tempfile
from pathlib import Path

import numpy as np
import pytest

REPO = Path(__file__).resolve().parent.parent

WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska
from spark_sklearn_amd.parallel import DistributedBackend

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(0)
n, f = 600, 16
X = rng.standard_normal((n, f))
y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)

backend = DistributedBackend()

# RC1 broadcast path: only rank 0 supplies the data
Xb, yb = backend.broadcast_data(X if rank == 0 else None,
                                y if rank == 0 else None)
assert Xb.shape == (n, f)
Xb, yb = Xb.numpy(), yb.numpy()
if rank == 0:
    assert np.allclose(Xb, X)

from spark_sklearn_amd.models import LogisticRegression
gs = ska.GridSearchCV("dist", LogisticRegression(max_iter=40),
                      {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3)
gs.fit(Xb, yb)

# every rank must hold identical aggregated results + a usable refit model
pred = gs.predict(Xb[:20])
out = {
    "rank": rank,
    "mean_test_score": gs.cv_results_["mean_test_score"].tolist(),
    "rank_test_score": gs.cv_results_["rank_test_score"].tolist(),
    "best_params": gs.best_params_,
    "best_score": gs.best_score_,
    "pred": np.asarray(pred).tolist(),
}
print("RESULT" + json.dumps(out))
dist.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_gloo_world2_search_matches_single_process(tmp_path):
    init_file = tmp_path / "init"
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ,
               SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}",
               WORLD_SIZE="2",
               MASTER_ADDR="127.0.0.1")
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=240)
        assert p.returncode == 0, f"worker failed:\n{stdout}\n{stderr}"
        line = [ln for ln in stdout.splitlines() if ln.startswith("RESULT")][0]
        outs.append(json.loads(line[len("RESULT"):]))

    r0, r1 = sorted(outs, key=lambda o: o["rank"])
    # both ranks agree bit-for-bit (RC2 all-reduce gives everyone the table)
    assert r0["mean_test_score"] == r1["mean_test_score"]
    assert r0["rank_test_score"] == r1["rank_test_score"]
    assert r0["best_params"] == r1["best_params"]
    assert r0["pred"] == r1["pred"]      # rank-0 refit broadcast worked

    # and the distributed result matches a single-process run exactly
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = 600, 16
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)
    gs = ska.GridSearchCV("sequential", LogisticRegression(max_iter=40),
                          {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3).fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               r0["mean_test_score"], rtol=1e-10)
    assert gs.best_params_ == r0["best_params"]


MULTI_WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(1)
n, f = 400, 12
X = rng.standard_normal((n, f))
y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)

from sklearn.svm import LinearSVC
gs = ska.GridSearchCV("dist", LinearSVC(),
                      {"C": [0.01, 0.1, 1.0]}, cv=3,
                      scoring={"acc": "accuracy",
                               "ba": "balanced_accuracy"},
                      refit="acc")
gs.fit(X, y)
out = {
    "rank": rank,
    "mean_acc": gs.cv_results_["mean_test_acc"].tolist(),
    "mean_ba": gs.cv_results_["mean_test_ba"].tolist(),
    "rank_acc": gs.cv_results_["rank_test_acc"].tolist(),
    "best_params": gs.best_params_,
    "best_score": gs.best_score_,
}
print("RESULT" + json.dumps(out))
dist.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_gloo_world2_multimetric(tmp_path):
    """RC2 with widened per-metric score columns: both ranks agree and
    match a single-process multimetric run."""
    init_file = tmp_path / "init"
    script = tmp_path / "worker.py"
    script.write_text(MULTI_WORKER)
    env = dict(os.environ,
               SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}",
               WORLD_SIZE="2",
               MASTER_ADDR="127.0.0.1")
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=240)
        assert p.returncode == 0, f"worker failed:\n{stdout}\n{stderr}"
        line = [ln for ln in stdout.splitlines()
                if ln.startswith("RESULT")][0]
        outs.append(json.loads(line[len("RESULT"):]))

    r0, r1 = sorted(outs, key=lambda o: o["rank"])
    assert r0["mean_acc"] == r1["mean_acc"]
    assert r0["mean_ba"] == r1["mean_ba"]
    assert r0["best_params"] == r1["best_params"]

    import spark_sklearn_amd as ska
    from sklearn.svm import LinearSVC

    rng = np.random.default_rng(1)
    n, f = 400, 12
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)
    gs = ska.GridSearchCV("sequential", LinearSVC(),
                          {"C": [0.01, 0.1, 1.0]}, cv=3,
                          scoring={"acc": "accuracy",
                                   "ba": "balanced_accuracy"},
                          refit="acc").fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_acc"],
                               r0["mean_acc"], rtol=1e-10)
    np.testing.assert_allclose(gs.cv_results_["mean_test_ba"],
                               r0["mean_ba"], rtol=1e-10)
    assert gs.best_params_ == r0["best_params"]


RESUME_WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska
from spark_sklearn_amd import HipContext

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(2)
n, f = 300, 10
X = rng.standard_normal((n, f))
y = (X[:, 0] > 0).astype(np.int64)

from sklearn.svm import LinearSVC
log = os.environ["SKA_LOG"]
# phase 1 (rank 0 already wrote a partial log before launch); phase 2:
# resume with sparse surviving task indices through the RC2 table
ctx = HipContext(backend="dist", resume_from_log=log, score_log=log)
gs = ska.GridSearchCV(ctx, LinearSVC(),
                      {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3)
gs.fit(X, y)
out = {"rank": rank, "resumed": gs.n_resumed_, "ran": gs.n_fits_,
       "mean": gs.cv_results_["mean_test_score"].tolist(),
       "best": gs.best_params_}
print("RESULT" + json.dumps(out))
dist.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_gloo_world2_resume_sparse_indices(tmp_path):
    """Resume + distributed: surviving task indices are sparse; the RC2
    table must index rows positionally (regression for an index-overflow
    bug)."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd import HipContext
    from sklearn.svm import LinearSVC

    rng = np.random.default_rng(2)
    n, f = 300, 10
    X = rng.standard_normal((n, f))
    y = (X[:, 0] > 0).astype(np.int64)
    log = tmp_path / "scores.jsonl"

    # seed the log with 2 of the 4 candidates (single process)
    ska.GridSearchCV(HipContext(backend="sequential",
                                score_log=str(log)),
                     LinearSVC(), {"C": [0.1, 1.0]}, cv=3).fit(X, y)

    init_file = tmp_path / "init"
    script = tmp_path / "worker.py"
    script.write_text(RESUME_WORKER)
    env = dict(os.environ, SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}", WORLD_SIZE="2",
               MASTER_ADDR="127.0.0.1", SKA_LOG=str(log))
    procs = [subprocess.Popen([sys.executable, str(script)],
                              env=dict(env, RANK=str(r)),
                              stdout=subprocess.PIPE,
                              stderr=subprocess.PIPE, text=True)
             for r in range(2)]
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=240)
        assert p.returncode == 0, f"worker failed:\n{stdout}\n{stderr}"
        line = [ln for ln in stdout.splitlines()
                if ln.startswith("RESULT")][0]
        outs.append(json.loads(line[len("RESULT"):]))
    r0, r1 = sorted(outs, key=lambda o: o["rank"])
    assert r0["resumed"] == 6 and r0["ran"] == 6
    assert r0["mean"] == r1["mean"]

    ref = ska.GridSearchCV("sequential", LinearSVC(),
                           {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3).fit(X, y)
    np.testing.assert_allclose(ref.cv_results_["mean_test_score"],
                               r0["mean"], rtol=1e-10)
