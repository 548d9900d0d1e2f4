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


SW_WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(2)
n, f = 500, 12
X = rng.standard_normal((n, f))
y = (X[:, 0] + 0.2 * rng.standard_normal(n) > 0).astype(np.int64)
w = np.where(y == 1, 5.0, 1.0)     # passed on EVERY rank (see docs/api.md)

from spark_sklearn_amd.models import LogisticRegression
gs = ska.GridSearchCV("dist", LogisticRegression(max_iter=25),
                      {"C": [0.1, 1.0]}, cv=3)
gs.fit(X, y, sample_weight=w)
out = {
    "rank": rank,
    "mean_test_score": gs.cv_results_["mean_test_score"].tolist(),
    "best_params": gs.best_params_,
}
print("RESULT" + json.dumps(out))
dist.barrier()
dist.destroy_process_group()
"""


@pytest.mark.timeout(300)
def test_gloo_world2_sample_weight(tmp_path):
    """Weighted search through the distributed tier: both ranks agree,
    and the result matches a weighted single-process run exactly."""
    init_file = tmp_path / "init"
    script = tmp_path / "worker_sw.py"
    script.write_text(SW_WORKER)
    env = dict(os.environ, SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}", WORLD_SIZE="2",
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
    assert r0["mean_test_score"] == r1["mean_test_score"]
    assert r0["best_params"] == r1["best_params"]

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(2)
    n, f = 500, 12
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.2 * rng.standard_normal(n) > 0).astype(np.int64)
    w = np.where(y == 1, 5.0, 1.0)
    gs = ska.GridSearchCV("sequential", LogisticRegression(max_iter=25),
                          {"C": [0.1, 1.0]}, cv=3).fit(X, y,
                                                       sample_weight=w)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               r0["mean_test_score"], rtol=1e-10)


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


BCAST_WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

# ONLY rank 0 constructs the data; fit() must broadcast it (RC1 inside
# fit -- the reference's one-line usage pattern survives)
if rank == 0:
    rng = np.random.default_rng(0)
    n, f = 500, 12
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)
else:
    X, y = None, None

from spark_sklearn_amd.models import LogisticRegression
gs = ska.GridSearchCV("dist", LogisticRegression(max_iter=30),
                      {"C": [0.1, 1.0]}, cv=3)
gs.fit(X, y)
print("RESULT" + json.dumps({
    "rank": rank,
    "mean_test_score": list(gs.cv_results_["mean_test_score"]),
    "best_params": gs.best_params_,
}))
dist.barrier()
dist.destroy_process_group()
"""


ERRTEXT_WORKER = r"""
import json, os, sys, warnings
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(1)
X = rng.standard_normal((300, 8))
y = (X[:, 0] > 0).astype(np.int64)

# candidate 1 fails (C<0); with candidate-round-robin it runs on rank 1,
# so rank 0 only sees its error text through the RC2 object gather
from sklearn.svm import LinearSVC
gs = ska.GridSearchCV("dist", LinearSVC(max_iter=30, dual="auto"),
                      {"C": [1.0, -5.0]}, cv=2, error_score=0.0)
with warnings.catch_warnings(record=True) as w:
    warnings.simplefilter("always")
    gs.fit(X, y)
texts = "\n".join(str(x.message) for x in w)
print("RESULT" + json.dumps({
    "rank": rank,
    "scores": list(gs.cv_results_["mean_test_score"]),
    "has_real_error_text": ("C" in texts and
                            "not gathered" not in texts and
                            len(texts) > 0),
}))
dist.barrier()
dist.destroy_process_group()
"""


DYN_WORKER = r"""
import json, os, sys, time
sys.path.insert(0, os.environ["SKA_REPO"])
import numpy as np
import torch.distributed as dist
import spark_sklearn_amd as ska
from spark_sklearn_amd import HipContext

rank = int(os.environ["RANK"])
dist.init_process_group("gloo", init_method=os.environ["SKA_INIT"],
                        rank=rank, world_size=int(os.environ["WORLD_SIZE"]))

rng = np.random.default_rng(0)
X = rng.standard_normal((400, 8))
y = (X[:, 0] > 0).astype(np.int64)

# skewed per-candidate cost: high max_iter candidates are ~8x the work.
# Static round-robin would put ALL the expensive candidates on one rank
# (they alternate in the grid); the dynamic queue rebalances.
from spark_sklearn_amd.models import LogisticRegression
ctx = HipContext(backend="dist", partition="dynamic", dynamic_block=1)
# rank 1 is slow per block (simulates skewed per-candidate cost); the
# dynamic queue should shift most blocks to rank 0
if rank == 1:
    import spark_sklearn_amd.parallel.distributed as D
    _orig = D.execute_local_tasks
    def _slow(*a, **k):
        time.sleep(0.3)
        return _orig(*a, **k)
    D.execute_local_tasks = _slow
gs = ska.GridSearchCV(ctx, LogisticRegression(max_iter=60),
                      {"C": [0.01, 0.1, 1.0, 10.0, 100.0, 1000.0]}, cv=2)
gs.fit(X, y)
pulled = getattr(ctx.backend, "n_blocks_pulled_", -1)
print("RESULT" + json.dumps({
    "rank": rank,
    "pulled": pulled,
    "mean_test_score": list(gs.cv_results_["mean_test_score"]),
    "best_params": gs.best_params_,
}))
dist.barrier()
dist.destroy_process_group()
"""


def _run2(tmp_path, worker_src, timeout=240):
    init_file = tmp_path / "init"
    script = tmp_path / "worker.py"
    script.write_text(worker_src)
    env = dict(os.environ, SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}", WORLD_SIZE="2",
               MASTER_ADDR="127.0.0.1")
    procs = [subprocess.Popen([sys.executable, str(script)],
                              env=dict(env, RANK=str(r)),
                              stdout=subprocess.PIPE,
                              stderr=subprocess.PIPE, text=True)
             for r in range(2)]
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=timeout)
        assert p.returncode == 0, f"worker failed:\n{stdout}\n{stderr}"
        line = [ln for ln in stdout.splitlines()
                if ln.startswith("RESULT")][0]
        outs.append(json.loads(line[len("RESULT"):]))
    return sorted(outs, key=lambda o: o["rank"])


@pytest.mark.timeout(300)
def test_gloo_world4_search_matches_single_process(tmp_path):
    """World size 4 (half the driver's 8-GPU shape): candidate
    round-robin across 4 ranks, identical aggregated results on every
    rank, equal to a single-process run."""
    init_file = tmp_path / "init"
    script = tmp_path / "worker4.py"
    script.write_text(WORKER)
    env = dict(os.environ, SKA_REPO=str(REPO),
               SKA_INIT=f"file://{init_file}", WORLD_SIZE="4",
               MASTER_ADDR="127.0.0.1")
    procs = []
    for rank in range(4):
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
    outs.sort(key=lambda o: o["rank"])
    for o in outs[1:]:
        assert o["mean_test_score"] == outs[0]["mean_test_score"]
        assert o["best_params"] == outs[0]["best_params"]
        assert o["pred"] == outs[0]["pred"]

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = 600, 16
    X = rng.standard_normal((n, f))
    y = (X[:, 0] + 0.1 * rng.standard_normal(n) > 0).astype(np.int64)
    gs = ska.GridSearchCV("sequential", LogisticRegression(max_iter=40),
                          {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3).fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               outs[0]["mean_test_score"], rtol=1e-10)


def test_gloo_world2_fit_broadcasts_rank0_data(tmp_path):
    """VERDICT round-1 item 3: fit() owns the RC1 broadcast."""
    r0, r1 = _run2(tmp_path, BCAST_WORKER)
    assert r0["mean_test_score"] == r1["mean_test_score"]
    assert r0["best_params"] == r1["best_params"]
    assert len(r0["mean_test_score"]) == 2


def test_gloo_world2_remote_error_text_gathered(tmp_path):
    """VERDICT round-1 item 4: FitFailedWarning keeps its diagnostic
    payload when the failing fit ran on another rank."""
    r0, r1 = _run2(tmp_path, ERRTEXT_WORKER)
    assert r0["scores"] == r1["scores"]
    assert r0["scores"][1] == 0.0            # failed candidate
    assert r0["has_real_error_text"]
    assert r1["has_real_error_text"]


def test_gloo_world2_dynamic_partition_balances(tmp_path):
    """Dynamic candidate-block queue: a late-starting rank pulls fewer
    blocks, results identical to the static single-process run."""
    r0, r1 = _run2(tmp_path, DYN_WORKER)
    assert r0["mean_test_score"] == r1["mean_test_score"]
    assert r0["pulled"] + r1["pulled"] == 6   # all 6 blocks processed once
    assert r0["pulled"] > r1["pulled"]        # rank 0 absorbed the skew
    # matches a single-process run
    import spark_sklearn_amd as ska
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    X = rng.standard_normal((400, 8))
    y = (X[:, 0] > 0).astype(np.int64)
    gs = ska.GridSearchCV("sequential", LogisticRegression(max_iter=60),
                          {"C": [0.01, 0.1, 1.0, 10.0, 100.0, 1000.0]},
                          cv=2).fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               r0["mean_test_score"], rtol=1e-10)
