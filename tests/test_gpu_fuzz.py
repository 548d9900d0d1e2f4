"""GPU kernel-tier shape fuzzing.

Random (rows, features, classes, candidates) shapes through the DEVICE
executors, compared against the host tier on identical data and plans --
the padding/staging coverage class of bug (e.g. unusual Fp x Bp tile
combinations) is exactly what fixed-shape tests miss.
"""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs a GPU", allow_module_level=True)

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from sklearn.model_selection import KFold  # noqa: E402

from spark_sklearn_amd.base_search import Task  # noqa: E402
from spark_sklearn_amd.engine import run_batched_tasks  # noqa: E402


def _problem(n, f, n_classes, seed):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal((f, n_classes))
    y = np.argmax(X @ w + 0.3 * rng.standard_normal((n, n_classes)),
                  axis=1)
    return X, y


def _run(est, X, y, cands, device):
    splits = list(KFold(2).split(np.arange(len(X))))
    tasks = []
    i = 0
    for ci, p in enumerate(cands):
        for fi, (tr, te) in enumerate(splits):
            tasks.append(Task(i, ci, fi, p, tr, te))
            i += 1
    out = run_batched_tasks(
        est, X, y, tasks, scorer=None, fit_params=None,
        return_train_score=True, error_score="raise", verbose=0,
        device=device)
    return {r.index: (r.test_score, r.train_score) for r in out}


@settings(max_examples=8, deadline=None)
@given(n=st.integers(300, 4000), f=st.integers(3, 200),
       n_classes=st.integers(2, 5), seed=st.integers(0, 10_000))
def test_fuzz_device_linear_matches_host(n, f, n_classes, seed):
    from spark_sklearn_amd.models import LogisticRegression

    X, y = _problem(n, f, n_classes, seed)
    if len(np.unique(y)) < n_classes:
        return
    est = LogisticRegression(max_iter=6, tol=0.0)
    cands = [{"C": 0.3}, {"C": 3.0}, {"C": 30.0}]
    dev = _run(est, X, y, cands, torch.device("cuda", 0))
    host = _run(est, X, y, cands, None)
    assert set(dev) == set(host)
    for idx in dev:
        # bf16 storage vs fp32 host: scores agree to bf16-level tolerance
        assert abs(dev[idx][0] - host[idx][0]) < 0.03, \
            (n, f, n_classes, idx, dev[idx], host[idx])
        assert abs(dev[idx][1] - host[idx][1]) < 0.03


@settings(max_examples=6, deadline=None)
@given(n=st.integers(500, 6000), f=st.integers(2, 60),
       depth=st.integers(1, 4), seed=st.integers(0, 10_000))
def test_fuzz_device_gbt_matches_host(n, f, depth, seed):
    from spark_sklearn_amd.models import GradientBoostingClassifier

    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = ((X[:, 0] + 0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
    est = GradientBoostingClassifier(n_estimators=5, max_depth=depth,
                                     learning_rate=0.3)
    cands = [{"max_depth": depth}]
    dev = _run(est, X, y, cands, torch.device("cuda", 0))
    host = _run(est, X, y, cands, None)
    for idx in dev:
        assert abs(dev[idx][0] - host[idx][0]) < 0.02, \
            (n, f, depth, idx, dev[idx], host[idx])


@settings(max_examples=6, deadline=None)
@given(n=st.integers(500, 6000), f=st.integers(2, 100),
       k=st.integers(2, 24), seed=st.integers(0, 10_000))
def test_fuzz_device_kmeans_matches_host(n, f, k, seed):
    from spark_sklearn_amd.models import KMeans

    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    est = KMeans(n_clusters=k, n_init=1, max_iter=8, random_state=0)
    cands = [{"n_clusters": k}]
    dev = _run(est, X, None, cands, torch.device("cuda", 0))
    host = _run(est, X, None, cands, None)
    for idx in dev:
        # negative inertia; bf16 distance arithmetic on device
        rel = abs(dev[idx][0] - host[idx][0]) / max(abs(host[idx][0]),
                                                    1e-9)
        assert rel < 0.02, (n, f, k, idx, dev[idx], host[idx])


@pytest.mark.parametrize("f", [129, 192, 300])
def test_odd_feature_widths_regression(f):
    """Pinned fuzz findings: feature counts padding past 128 must route
    through a kernel-supported Fp (129->256, 300->512)."""
    from spark_sklearn_amd.models import LogisticRegression

    X, y = _problem(2000, f, 3, seed=1)
    est = LogisticRegression(max_iter=6, tol=0.0)
    cands = [{"C": 1.0}]
    dev = _run(est, X, y, cands, torch.device("cuda", 0))
    host = _run(est, X, y, cands, None)
    for idx in dev:
        assert abs(dev[idx][0] - host[idx][0]) < 0.03
