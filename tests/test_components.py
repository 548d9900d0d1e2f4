"""Parity-component tests: Converter, KeyedEstimator/KeyedModel, gapply,
CSRVectorUDT (reference test files test_converter.py / test_keyed_models.py
/ test_gapply.py)."""

import numpy as np
import pandas as pd
import pytest
import scipy.sparse as sp
from sklearn.cluster import KMeans
from sklearn.linear_model import LogisticRegression as SKLogReg
from sklearn.preprocessing import StandardScaler

from spark_sklearn_amd import (Converter, CSRVectorUDT, KeyedEstimator,
                               SparkSklearnEstimator, gapply)
from spark_sklearn_amd.models import LinearRegression, LogisticRegression
from spark_sklearn_amd.udt import DeviceCSRMatrix, stack_csr_rows


# ---------------------------- Converter -----------------------------------

def test_converter_roundtrip_logreg(binary_blobs):
    X, y = binary_blobs
    ours = LogisticRegression(max_iter=100).fit(X, y)
    conv = Converter()
    sk = conv.toSKLearn(ours)
    assert isinstance(sk, SKLogReg)
    # predictions must agree (same linear decision function)
    np.testing.assert_array_equal(sk.predict(X), ours.predict(X))
    back = conv.toHip(sk)
    np.testing.assert_allclose(back.coef_, ours.coef_, rtol=1e-12)
    np.testing.assert_array_equal(back.predict(X), ours.predict(X))


def test_converter_linear_regression():
    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 5))
    y = X @ rng.standard_normal(5) + 1.0
    ours = LinearRegression(max_iter=400).fit(X, y)
    sk = Converter().toSKLearn(ours)
    np.testing.assert_allclose(sk.predict(X), ours.predict(X), rtol=1e-5)


def test_converter_rejects_unsupported():
    with pytest.raises(ValueError):
        Converter().toSKLearn(KMeans())
    with pytest.raises(ValueError):
        Converter().toHip(StandardScaler())


def test_converter_unfitted_raises(binary_blobs):
    with pytest.raises(ValueError, match="not fitted"):
        Converter().toSKLearn(LogisticRegression())


# ------------------------- KeyedEstimator ---------------------------------

def _keyed_df():
    rng = np.random.default_rng(0)
    rows = []
    for k in range(3):
        center = np.array([k * 10.0, -k * 5.0])
        for _ in range(30):
            x = center + rng.standard_normal(2)
            rows.append({"k": k, "features": x, "label": float(x[0] > k * 10)})
    return pd.DataFrame(rows)


def test_keyed_clusterer():
    df = _keyed_df()
    ke = KeyedEstimator(sklearnEstimator=KMeans(n_clusters=2, n_init=2,
                                                random_state=0),
                        keyCols=["k"])
    model = ke.fit(df)
    assert ke.estimatorType == "clusterer"
    assert len(model.keyedModels) == 3
    out = model.transform(df)
    assert "prediction" in out.columns
    assert len(out) == len(df)
    assert list(out.index) == list(df.index)


def test_keyed_predictor_matches_per_key_sklearn():
    df = _keyed_df()
    ke = KeyedEstimator(sklearnEstimator=SKLogReg(max_iter=200),
                        keyCols=["k"], xCol="features", yCol="label")
    model = ke.fit(df)
    assert ke.estimatorType == "predictor"
    out = model.transform(df)
    # oracle: per-key local sklearn fit must give identical predictions
    for k, group in df.groupby("k"):
        X = np.stack(group["features"].values)
        y = group["label"].values
        ref = SKLogReg(max_iter=200).fit(X, y)
        got = out.loc[group.index, "prediction"].values
        np.testing.assert_array_equal(got, ref.predict(X))


def test_keyed_transformer():
    df = _keyed_df()
    ke = KeyedEstimator(sklearnEstimator=StandardScaler(), keyCols=["k"])
    model = ke.fit(df)
    assert ke.estimatorType == "transformer"
    out = model.transform(df)
    assert "output" in out.columns
    for k, group in out.groupby("k"):
        vals = np.stack(group["output"].values)
        np.testing.assert_allclose(vals.mean(axis=0), 0.0, atol=1e-9)


def test_keyed_unseen_key():
    df = _keyed_df()
    model = KeyedEstimator(sklearnEstimator=KMeans(n_clusters=2, n_init=2),
                           keyCols=["k"]).fit(df)
    extra = pd.DataFrame(
        [{"k": 99, "features": np.zeros(2)}])
    with pytest.raises(KeyError):
        model.transform(extra)
    dropped = model.transform(extra, unseenKeys="drop")
    assert len(dropped) == 0


def test_keyed_validation():
    with pytest.raises(ValueError):
        KeyedEstimator(sklearnEstimator=None)
    with pytest.raises(ValueError):
        KeyedEstimator(sklearnEstimator=KMeans(), estimatorType="predictor")
    with pytest.raises(ValueError):
        KeyedEstimator(sklearnEstimator=KMeans(),
                       estimatorType="bogus")
    wrapper = SparkSklearnEstimator(KMeans(n_clusters=4))
    assert wrapper.n_clusters == 4


# ------------------------------ gapply ------------------------------------

def test_gapply_matches_pandas_groupby():
    df = pd.DataFrame({"g": [0, 0, 1, 1, 1], "v": [1.0, 2.0, 3.0, 4.0, 5.0],
                       "w": [10.0, 20.0, 30.0, 40.0, 50.0]})

    def summarize(key, pdf):
        return pd.DataFrame({"v_sum": [pdf["v"].sum()],
                             "w_mean": [pdf["w"].mean()]})

    out = gapply((df, "g"), summarize, ["v_sum", "w_mean"], "v", "w")
    oracle = df.groupby("g").agg(v_sum=("v", "sum"), w_mean=("w", "mean"))
    np.testing.assert_allclose(out["v_sum"].values, oracle["v_sum"].values)
    np.testing.assert_allclose(out["w_mean"].values, oracle["w_mean"].values)
    assert list(out["g"]) == [0, 1]


def test_gapply_schema_mismatch_and_empty():
    df = pd.DataFrame({"g": [0, 1], "v": [1.0, 2.0]})
    with pytest.raises(ValueError, match="schema"):
        gapply((df, "g"), lambda k, p: pd.DataFrame({"x": [1]}),
               ["wrong"], "v")
    out = gapply((df, "g"), lambda k, p: None, ["x"], "v")
    assert len(out) == 0 and list(out.columns) == ["g", "x"]


def test_gapply_no_retain_group_columns():
    df = pd.DataFrame({"g": [0, 0, 1], "v": [1.0, 2.0, 3.0]})
    out = gapply((df, "g"), lambda k, p: pd.DataFrame({"s": [p["v"].sum()]}),
                 ["s"], "v", retainGroupColumns=False)
    assert list(out.columns) == ["s"]
    assert sorted(out["s"]) == [3.0, 3.0]


# ------------------------------- UDT --------------------------------------

def test_csr_vector_udt_roundtrip():
    row = sp.csr_matrix(np.array([[0.0, 1.5, 0.0, -2.0]]))
    rec = CSRVectorUDT.serialize(row)
    assert rec["size"] == 4
    back = CSRVectorUDT.deserialize(rec)
    np.testing.assert_allclose(back.toarray(), row.toarray())
    with pytest.raises(ValueError):
        CSRVectorUDT.serialize(sp.csr_matrix(np.eye(2)))


def test_stack_csr_rows_mixed():
    rows = [sp.csr_matrix(np.array([[1.0, 0.0]])),
            {"indices": np.array([1]), "values": np.array([2.0]), "size": 2},
            np.array([3.0, 4.0])]
    out = stack_csr_rows(rows)
    np.testing.assert_allclose(out.toarray(),
                               [[1, 0], [0, 2], [3, 4]])


def test_device_csr_cpu_roundtrip():
    mat = sp.random(50, 20, density=0.1, format="csr", random_state=0)
    dev = DeviceCSRMatrix.from_scipy(mat, device="cpu")
    back = dev.to_scipy()
    assert back.shape == mat.shape
    assert dev.nnz == mat.nnz
    # bf16 storage tolerance
    np.testing.assert_allclose(back.toarray(), mat.toarray(),
                               atol=0.01, rtol=0.01)


def test_score_log_resume(tmp_path):
    """A killed sweep's JSONL score log replays on the next fit: completed
    (candidate, fold) fits are restored, only the remainder executes, and
    cv_results_ matches an uninterrupted run exactly."""
    import numpy as np

    import spark_sklearn_amd as ska
    from spark_sklearn_amd import HipContext
    from sklearn.svm import LinearSVC

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 8))
    y = (X[:, 0] > 0).astype(np.int64)
    log = str(tmp_path / "scores.jsonl")
    grid = {"C": [0.01, 0.1, 1.0, 10.0]}

    # phase 1: "killed" run that only saw half the grid
    ctx1 = HipContext(backend="sequential", score_log=log)
    ska.GridSearchCV(ctx1, LinearSVC(), {"C": [0.01, 0.1]}, cv=3).fit(X, y)

    # phase 2: full grid, resuming -- only the 2 new candidates run
    ctx2 = HipContext(backend="sequential", score_log=log,
                      resume_from_log=log)
    gs = ska.GridSearchCV(ctx2, LinearSVC(), grid, cv=3)
    gs.fit(X, y)
    assert gs.n_resumed_ == 6          # 2 candidates x 3 folds restored
    assert gs.n_fits_ == 6             # 2 new candidates x 3 folds run

    ref = ska.GridSearchCV("sequential", LinearSVC(), grid, cv=3).fit(X, y)
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               ref.cv_results_["mean_test_score"],
                               rtol=1e-12)
    assert gs.best_params_ == ref.best_params_

    # phase 3: everything restored, zero fits executed
    gs2 = ska.GridSearchCV(HipContext(backend="sequential",
                                      resume_from_log=log),
                           LinearSVC(), grid, cv=3).fit(X, y)
    assert gs2.n_fits_ == 0 and gs2.n_resumed_ == 12
    np.testing.assert_allclose(gs2.cv_results_["mean_test_score"],
                               ref.cv_results_["mean_test_score"],
                               rtol=1e-12)


def test_progress_callback():
    """HipContext(progress=fn) receives (done, total) ticks ending at
    (total, total); a raising callback must not break the search."""
    import numpy as np

    import spark_sklearn_amd as ska
    from spark_sklearn_amd import HipContext
    from sklearn.svm import LinearSVC

    rng = np.random.default_rng(0)
    X = rng.standard_normal((200, 6))
    y = (X[:, 0] > 0).astype(np.int64)

    calls = []
    ctx = HipContext(backend="sequential",
                     progress=lambda d, t: calls.append((d, t)))
    ska.GridSearchCV(ctx, LinearSVC(), {"C": [0.1, 1.0]}, cv=3).fit(X, y)
    assert calls and calls[-1] == (6, 6)
    assert all(t == 6 and 1 <= d <= 6 for d, t in calls)

    def boom(d, t):
        raise RuntimeError("callback bug")

    ctx2 = HipContext(backend="sequential", progress=boom)
    gs = ska.GridSearchCV(ctx2, LinearSVC(), {"C": [1.0]}, cv=3).fit(X, y)
    assert hasattr(gs, "best_score_")


def test_hip_error_retry(monkeypatch):
    """A transient HIP RuntimeError on a candidate chunk retries once
    (SURVEY.md section 5.3a); persistent failure propagates."""
    import numpy as np

    from sklearn.model_selection import KFold

    from spark_sklearn_amd.base_search import Task
    from spark_sklearn_amd.engine import (LinearFamily, run_batched_tasks,
                                          _HostLinearExecutor)
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    X = rng.standard_normal((200, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)

    fails = {"n": 1}
    orig_fit = _HostLinearExecutor.fit

    def flaky_fit(self, train_idx, plan):
        if fails["n"] > 0:
            fails["n"] -= 1
            raise RuntimeError("HIP error: out of resources (fake)")
        return orig_fit(self, train_idx, plan)

    monkeypatch.setattr(_HostLinearExecutor, "fit", flaky_fit)
    splits = list(KFold(3).split(np.arange(len(X))))
    tasks = [Task(i, 0, f, {"C": 1.0}, tr, te)
             for i, (f, (tr, te)) in enumerate(enumerate(splits))]
    with pytest.warns(RuntimeWarning, match="retrying candidate chunk"):
        results = run_batched_tasks(
            LogisticRegression(max_iter=5), X, y, tasks, scorer=None,
            fit_params=None, return_train_score=False,
            error_score="raise", verbose=0, device=None)
    assert len(results) == 3 and all(r.error is None for r in results)


def test_prometheus_metrics():
    """Optional observability endpoint."""
    pytest.importorskip("prometheus_client")
    import socket
    import urllib.request

    import numpy as np

    import spark_sklearn_amd as ska
    from spark_sklearn_amd.util import enable_prometheus
    from sklearn.svm import LinearSVC

    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    enable_prometheus(port)

    rng = np.random.default_rng(0)
    X = rng.standard_normal((150, 5))
    y = (X[:, 0] > 0).astype(np.int64)
    ska.GridSearchCV("sequential", LinearSVC(), {"C": [0.1, 1.0]},
                     cv=3).fit(X, y)
    body = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
    assert "ska_candidate_fits_total 6.0" in body
    assert "ska_searches_total 1.0" in body
    assert "ska_fits_per_sec" in body


def test_candidate_batch_knob():
    """HipContext(candidate_batch=N) reaches the engine's chunking."""
    import numpy as np

    import spark_sklearn_amd as ska
    from spark_sklearn_amd import HipContext
    from spark_sklearn_amd.models import LogisticRegression
    import spark_sklearn_amd.engine as eng

    seen = []
    orig = eng.run_batched_tasks

    def spy(*a, **k):
        seen.append(k.get("column_batch"))
        return orig(*a, **k)

    rng = np.random.default_rng(0)
    X = rng.standard_normal((200, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)
    import spark_sklearn_amd.parallel.backend as pb
    old = eng.run_batched_tasks
    eng.run_batched_tasks = spy
    # backend module imported the symbol lazily inside the function, so
    # patching the engine module attribute is enough
    try:
        ctx = HipContext(backend="sequential", candidate_batch=64)
        # force the engine path even on CPU via direct call contract
        from spark_sklearn_amd.parallel.backend import execute_local_tasks
        from spark_sklearn_amd.base_search import Task
        from sklearn.model_selection import KFold

        splits = list(KFold(2).split(np.arange(len(X))))
        tasks = [Task(i, 0, f, {"C": 1.0}, tr, te)
                 for i, (f, (tr, te)) in enumerate(enumerate(splits))]
        import unittest.mock as mock
        with mock.patch.object(pb, "_device_available",
                               return_value=True):
            execute_local_tasks(
                LogisticRegression(max_iter=3), X, y, tasks, scorer=None,
                fit_params=None, return_train_score=False,
                error_score="raise", verbose=0, n_threads=1,
                device=None, column_batch=64)
    finally:
        eng.run_batched_tasks = old
    assert seen == [64]


def test_keyed_models_doctest():
    """The reference carried doctest examples in keyed_models; ours run."""
    import doctest

    import spark_sklearn_amd.keyed_models as km

    res = doctest.testmod(km)
    assert res.attempted >= 1 and res.failed == 0


def test_resume_log_header_mismatch_refused(tmp_path):
    """ADVICE round-1 (medium): a score log written against different
    data / cv / estimator must be refused, not silently replayed."""
    import spark_sklearn_amd as ska
    from spark_sklearn_amd import HipContext
    from spark_sklearn_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    X1 = rng.standard_normal((200, 6))
    y1 = (X1[:, 0] > 0).astype(np.int64)
    log = str(tmp_path / "scores.jsonl")
    ctx = HipContext(backend="sequential", score_log=log)
    ska.GridSearchCV(ctx, LogisticRegression(max_iter=20),
                     {"C": [0.1, 1.0]}, cv=2).fit(X1, y1)
    # header record present
    import json
    first = json.loads(open(log).readline())
    assert first.get("header") == 1 and "data" in first

    # same search + same data resumes cleanly (everything restored)
    ctx2 = HipContext(backend="sequential", resume_from_log=log)
    gs2 = ska.GridSearchCV(ctx2, LogisticRegression(max_iter=20),
                           {"C": [0.1, 1.0]}, cv=2)
    gs2.fit(X1, y1)
    assert gs2.n_resumed_ == 4

    # different data -> refused
    X2 = rng.standard_normal((200, 6))
    y2 = (X2[:, 1] > 0).astype(np.int64)
    ctx3 = HipContext(backend="sequential", resume_from_log=log)
    gs3 = ska.GridSearchCV(ctx3, LogisticRegression(max_iter=20),
                           {"C": [0.1, 1.0]}, cv=2)
    with pytest.raises(ValueError, match="resume_from_log refused"):
        gs3.fit(X2, y2)

    # different cv -> refused
    ctx4 = HipContext(backend="sequential", resume_from_log=log)
    gs4 = ska.GridSearchCV(ctx4, LogisticRegression(max_iter=20),
                           {"C": [0.1, 1.0]}, cv=3)
    with pytest.raises(ValueError, match="resume_from_log refused"):
        gs4.fit(X1, y1)

    # weighted rerun of an unweighted log -> refused (scores differ)
    ctx5 = HipContext(backend="sequential", resume_from_log=log)
    gs5 = ska.GridSearchCV(ctx5, LogisticRegression(max_iter=20),
                           {"C": [0.1, 1.0]}, cv=2)
    with pytest.raises(ValueError, match="resume_from_log refused"):
        gs5.fit(X1, y1, sample_weight=np.where(y1 == 1, 3.0, 1.0))

    # different CONSTRUCTOR params (not in the grid) -> refused
    ctx6 = HipContext(backend="sequential", resume_from_log=log)
    gs6 = ska.GridSearchCV(ctx6, LogisticRegression(max_iter=20,
                                                    momentum=0.5),
                           {"C": [0.1, 1.0]}, cv=2)
    with pytest.raises(ValueError, match="resume_from_log refused"):
        gs6.fit(X1, y1)


def test_keyed_estimator_with_rf():
    """KeyedEstimator's generic per-key path works with the RF family."""
    import pandas as pd

    from spark_sklearn_amd import KeyedEstimator
    from spark_sklearn_amd.models import RandomForestClassifier

    rng = np.random.default_rng(0)
    rows = []
    for key in ("a", "b"):
        X = rng.standard_normal((200, 6))
        y = (X[:, 0] > 0).astype(int)
        for i in range(200):
            rows.append({"k": key, "features": X[i], "label": y[i]})
    df = pd.DataFrame(rows)
    ke = KeyedEstimator(
        sklearnEstimator=RandomForestClassifier(n_estimators=5,
                                                max_depth=4,
                                                random_state=0),
        keyCols=["k"], xCol="features", yCol="label")
    model = ke.fit(df)
    out = model.transform(df)
    assert "prediction" in out.columns
    acc = (out["prediction"] == df["label"]).mean()
    assert acc > 0.8
