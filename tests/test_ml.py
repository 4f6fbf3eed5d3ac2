"""ML module tests (reference: bodo/ml_support tests pattern): fits on
distributed frames match sklearn-style closed forms."""

import numpy as np
import pandas as pd
import pytest
import torch

import bodo_amd.pandas as bpd
from bodo_amd.ml import LinearRegression, LogisticRegression, train_test_split


def _make_reg(n=5000, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(0, 1, (n, 4))
    w = np.array([1.5, -2.0, 0.5, 3.0])
    y = X @ w + 0.7 + rng.normal(0, 0.01, n)
    return X, y, w


def test_linreg_numpy():
    X, y, w = _make_reg()
    m = LinearRegression().fit(X, y)
    assert np.allclose(m.coef_, w, atol=1e-2)
    assert abs(m.intercept_ - 0.7) < 1e-2
    assert m.score(X, y) > 0.999


def test_linreg_frames():
    X, y, w = _make_reg(3000, 1)
    df = pd.DataFrame(X, columns=["x0", "x1", "x2", "x3"])
    df["y"] = y
    b = bpd.from_pandas(df)
    m = LinearRegression().fit(b[["x0", "x1", "x2", "x3"]], b["y"])
    assert np.allclose(m.coef_, w, atol=1e-2)


def test_logreg():
    rng = np.random.default_rng(2)
    n = 4000
    X = rng.normal(0, 1, (n, 3))
    z = X @ np.array([2.0, -1.0, 0.5]) + 0.3
    y = (z > 0).astype(np.float64)
    m = LogisticRegression(lr=1.0, max_iter=300).fit(X, y)
    acc = (m.predict(X).reshape(-1) == y).mean()
    assert acc > 0.95, acc


def test_train_test_split():
    X, y, _ = _make_reg(1000, 3)
    Xtr, Xte, ytr, yte = train_test_split(X, y, test_size=0.2, random_state=0)
    assert len(Xtr) == 800 and len(Xte) == 200


@pytest.mark.gpu
def test_mfma_gemm_numerics():
    """gemm_f32 (v_mfma_f32_16x16x4_f32) vs torch matmul fp64 reference."""
    import bodo_amd_kernels as K

    rng = np.random.default_rng(5)
    for (m, k, n) in [(64, 64, 64), (130, 96, 70), (256, 512, 128), (33, 7, 65)]:
        A = torch.from_numpy(rng.uniform(-1, 1, (m, k)).astype(np.float32)).cuda()
        B = torch.from_numpy(rng.uniform(-1, 1, (k, n)).astype(np.float32)).cuda()
        C = K.gemm_f32(A, B).cpu().numpy()
        ref = (A.cpu().double() @ B.cpu().double()).numpy()
        assert np.allclose(C, ref, atol=1e-4 * k ** 0.5), (m, k, n, np.abs(C - ref).max())


@pytest.mark.gpu
def test_linreg_gpu():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    try:
        X, y, w = _make_reg(20000, 7)
        m = LinearRegression().fit(X, y)
        assert np.allclose(m.coef_, w, atol=1e-2)
    finally:
        cfg.DEVICE = ""


def test_standard_scaler():
    from sklearn.preprocessing import StandardScaler as SkScaler

    from bodo_amd.ml import StandardScaler

    rng = np.random.default_rng(7)
    X = rng.random((500, 4)) * np.array([1.0, 10.0, 0.1, 5.0]) + 3.0
    ours = StandardScaler().fit(X)
    ref = SkScaler().fit(X)
    # inputs pass through the float32 device-matrix path
    np.testing.assert_allclose(ours.mean_, ref.mean_, rtol=1e-6)
    np.testing.assert_allclose(ours.scale_, ref.scale_, rtol=1e-5)
    np.testing.assert_allclose(ours.transform(X), ref.transform(X),
                               rtol=1e-4, atol=1e-5)


def test_kmeans_quality():
    from sklearn.cluster import KMeans as SkKMeans

    from bodo_amd.ml import KMeans

    rng = np.random.default_rng(8)
    centers = np.array([[0.0, 0.0], [10.0, 0.0], [0.0, 10.0], [10.0, 10.0]])
    X = np.concatenate([rng.normal(c, 0.5, size=(300, 2)) for c in centers])
    km = KMeans(n_clusters=4, random_state=3).fit(X.astype(np.float32))
    sk = SkKMeans(n_clusters=4, n_init=10, random_state=3).fit(X)
    # well-separated blobs: both must find the 4 true centers
    ours = np.sort(km.cluster_centers_.round(0), axis=0)
    theirs = np.sort(sk.cluster_centers_.round(0), axis=0)
    np.testing.assert_allclose(np.sort(ours.flatten()),
                               np.sort(theirs.flatten()), atol=1.0)
    assert km.inertia_ <= sk.inertia_ * 1.2
    pred = km.predict(X.astype(np.float32))
    assert len(np.unique(pred)) == 4


def test_preprocessing_and_metrics_vs_sklearn():
    """MinMaxScaler/LabelEncoder/train_test_split/metrics match sklearn
    single-rank (distributed partials all-reduce the same way)."""
    import sklearn.metrics as skmet
    import sklearn.preprocessing as skp

    from bodo_amd import ml

    rng = np.random.default_rng(0)
    X = rng.random((100, 3)) * 10 - 5
    y = rng.integers(0, 3, 100)
    mm = ml.MinMaxScaler().fit(X)
    np.testing.assert_allclose(mm.transform(X),
                               skp.MinMaxScaler().fit_transform(X),
                               rtol=1e-4)  # float32 device staging
    le = ml.LabelEncoder().fit(["b", "a", "c", "a"])
    assert le.transform(["a", "c"]).tolist() == [0, 2]
    assert le.inverse_transform([1]).tolist() == ["b"]
    Xtr, Xte, ytr, yte = ml.train_test_split(X, y, test_size=0.25,
                                             random_state=0)
    assert len(Xtr) == 75 and len(Xte) == 25 and len(ytr) == 75
    yp = (y + (rng.random(100) < 0.2)).clip(0, 2)
    assert abs(ml.accuracy_score(y, yp)
               - skmet.accuracy_score(y, yp)) < 1e-12
    assert abs(ml.mean_squared_error(y, yp)
               - skmet.mean_squared_error(y, yp)) < 1e-12
    assert abs(ml.r2_score(y, yp) - skmet.r2_score(y, yp)) < 1e-12
    assert abs(ml.mean_absolute_error(y, yp)
               - skmet.mean_absolute_error(y, yp)) < 1e-12
