"""KMeans family: reference solver numerics + search integration (CPU)."""

import numpy as np
import pytest
from sklearn.cluster import KMeans as SKKMeans

import spark_sklearn_amd as ska
from spark_sklearn_amd.models import KMeans
from spark_sklearn_amd.models._kmeans_ref import (KMeansPlan,
                                                  kmeans_fit_ref,
                                                  kmeans_init_centers)


@pytest.fixture(scope="module")
def blobs():
    rng = np.random.default_rng(0)
    centers = rng.standard_normal((4, 8)) * 4
    y = rng.integers(0, 4, 3000)
    X = centers[y] + rng.standard_normal((3000, 8))
    return X.astype(np.float64), y


def test_matches_sklearn_inertia(blobs):
    X, _ = blobs
    ours = KMeans(n_clusters=4, n_init=3, random_state=0, max_iter=100)
    ours.fit(X)
    sk = SKKMeans(n_clusters=4, n_init=3, random_state=0).fit(X)
    assert abs(ours.inertia_ - sk.inertia_) / sk.inertia_ < 0.02
    assert ours.cluster_centers_.shape == (4, 8)
    assert ours.labels_.shape == (3000,)
    # same partition up to label permutation
    from sklearn.metrics import adjusted_rand_score

    assert adjusted_rand_score(ours.labels_, sk.labels_) > 0.99
    # score() uses exact fp64 distances; inertia_ comes from the batched
    # solver whose distances are defined against bf16-rounded centers
    assert abs(ours.score(X) + ours.inertia_) < 1e-4 * ours.inertia_


def test_predict_transform(blobs):
    X, _ = blobs
    km = KMeans(n_clusters=4, random_state=0).fit(X)
    pred = km.predict(X[:50])
    assert set(pred) <= set(range(4))
    np.testing.assert_array_equal(pred, km.labels_[:50])
    T = km.transform(X[:10])
    assert T.shape == (10, 4)
    np.testing.assert_array_equal(np.argmin(T, axis=1), pred[:10])


def test_batched_blocks_equal_individual(blobs):
    """A 3-block batch must equal three separate 1-block runs."""
    X, _ = blobs
    X32 = X.astype(np.float32)
    rngs = [np.random.default_rng(s) for s in (1, 2, 3)]
    ks = [2, 4, 6]
    C0s = [kmeans_init_centers(X32, k, "k-means++", r)
           for k, r in zip(ks, rngs)]
    cstart = np.cumsum([0] + ks).astype(np.int32)
    plan = KMeansPlan(cstart=cstart,
                      block_to_cand=np.arange(3, dtype=np.int32),
                      max_iter=np.full(3, 50, np.int32),
                      tol_scaled=np.full(3, 1e-4, np.float32), n_cand=3)
    C_all, n_iter, inertia = kmeans_fit_ref(X32, np.concatenate(C0s), plan)
    for i, (k, C0) in enumerate(zip(ks, C0s)):
        p1 = KMeansPlan(cstart=np.array([0, k], np.int32),
                        block_to_cand=np.zeros(1, np.int32),
                        max_iter=np.full(1, 50, np.int32),
                        tol_scaled=np.full(1, 1e-4, np.float32), n_cand=1)
        C1, ni1, in1 = kmeans_fit_ref(X32, C0, p1)
        lo, hi = cstart[i], cstart[i + 1]
        np.testing.assert_allclose(C_all[lo:hi], C1, rtol=1e-5)
        assert n_iter[i] == ni1[0]
        np.testing.assert_allclose(inertia[i], in1[0], rtol=1e-6)


def test_kmeans_grid_search(blobs):
    X, _ = blobs
    gs = ska.GridSearchCV(None, KMeans(random_state=0, max_iter=60),
                          {"n_clusters": [2, 4, 8]}, cv=3,
                          return_train_score=False)
    gs.fit(X)
    assert len(gs.cv_results_["params"]) == 3
    # more clusters always lowers inertia -> rank follows k
    assert gs.best_params_["n_clusters"] == 8
    assert hasattr(gs.best_estimator_, "cluster_centers_")
    assert gs.best_estimator_.cluster_centers_.shape == (8, 8)


def test_kmeans_keyed_estimator(blobs):
    """Our KMeans works inside KeyedEstimator like any sklearn clusterer."""
    import pandas as pd

    X, _ = blobs
    df = pd.DataFrame({
        "k": np.repeat([0, 1], 100),
        "features": list(X[:200]),
    })
    model = ska.KeyedEstimator(
        sklearnEstimator=KMeans(n_clusters=2, random_state=0),
        keyCols=["k"]).fit(df)
    out = model.transform(df)
    assert "prediction" in out.columns


def test_empty_cluster_keeps_center():
    X = np.array([[0.0], [0.1], [10.0]], dtype=np.float32)
    C0 = np.array([[0.0], [5.0], [100.0]], dtype=np.float32)  # last empty
    plan = KMeansPlan(cstart=np.array([0, 3], np.int32),
                      block_to_cand=np.zeros(1, np.int32),
                      max_iter=np.full(1, 5, np.int32),
                      tol_scaled=np.zeros(1, np.float32), n_cand=1)
    C, _, _ = kmeans_fit_ref(X, C0, plan)
    assert C[2, 0] == 100.0   # untouched empty cluster


def test_invalid_params_error_score(blobs):
    X, _ = blobs
    gs = ska.GridSearchCV(None, KMeans(random_state=0),
                          {"n_clusters": [-1, 4]}, cv=3,
                          error_score=-1e18, return_train_score=False)
    from sklearn.exceptions import FitFailedWarning

    with pytest.warns(FitFailedWarning):
        gs.fit(X)
    assert gs.best_params_["n_clusters"] == 4
