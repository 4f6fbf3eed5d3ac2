"""Distributed ML on bodo_amd frames (reference: bodo/ml_support/sklearn_ext.py
fit + allreduce pattern, bodo/ai/train.py torch bridge).

Linear models fit on the MFMA GEMM path (csrc/gemm.hip,
v_mfma_f32_16x16x4_f32): each rank computes its shard's Gram matrix
X^T X and X^T y on device, partials are all-reduced over RCCL, and the
small normal-equations system is solved with torch.linalg.
"""

from __future__ import annotations

from typing import Optional, Union

import numpy as np
import pandas as pd
import torch

from ..parallel import comm


def _as_matrix(X, device) -> torch.Tensor:
    from ..pandas.frame import BodoDataFrame
    from ..pandas.series import BodoSeries

    if isinstance(X, BodoDataFrame):
        shard = X.execute()
        cols = [c.data.to(torch.float32) for c in shard.columns]
        return torch.stack(cols, dim=1).to(device)
    if isinstance(X, BodoSeries):
        ser = X.to_pandas()
        return torch.from_numpy(ser.to_numpy(dtype=np.float32)).reshape(-1, 1).to(device)
    if isinstance(X, pd.DataFrame):
        return torch.from_numpy(X.to_numpy(dtype=np.float32)).to(device)
    if isinstance(X, pd.Series):
        return torch.from_numpy(X.to_numpy(dtype=np.float32)).reshape(-1, 1).to(device)
    if isinstance(X, np.ndarray):
        return torch.from_numpy(np.ascontiguousarray(X, dtype=np.float32)).to(device)
    if torch.is_tensor(X):
        return X.to(device, torch.float32)
    raise TypeError(type(X))


def _matmul_at_b(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """A^T @ B via the MFMA kernel on GPU, torch on CPU."""
    if A.is_cuda:
        import bodo_amd_kernels as K

        return K.gemm_f32(A.t().contiguous(), B.contiguous())
    return A.t() @ B


def _allreduce_(t: torch.Tensor) -> torch.Tensor:
    if comm.initialized() and comm.get_world_size() > 1:
        import torch.distributed as dist

        wire = t
        moved = False
        if dist.get_backend() == "gloo" and t.is_cuda:
            wire = t.cpu()
            moved = True
        dist.all_reduce(wire)
        if moved:
            t.copy_(wire)
    return t


class LinearRegression:
    """Least squares via normal equations on the MFMA GEMM path."""

    def __init__(self, fit_intercept: bool = True, alpha: float = 0.0):
        self.fit_intercept = fit_intercept
        self.alpha = alpha  # ridge regularization
        self.coef_: Optional[np.ndarray] = None
        self.intercept_: float = 0.0

    def _device(self):
        from .. import config

        return torch.device(config.default_device())

    def fit(self, X, y):
        dev = self._device()
        Xm = _as_matrix(X, dev)
        ym = _as_matrix(y, dev).reshape(-1, 1)
        if self.fit_intercept:
            ones = torch.ones(Xm.shape[0], 1, dtype=torch.float32, device=dev)
            Xm = torch.cat([Xm, ones], dim=1)
        xtx = _matmul_at_b(Xm, Xm).to(torch.float64)
        xty = _matmul_at_b(Xm, ym).to(torch.float64)
        _allreduce_(xtx)
        _allreduce_(xty)
        if self.alpha:
            xtx += self.alpha * torch.eye(xtx.shape[0], dtype=torch.float64,
                                          device=xtx.device)
        w = torch.linalg.solve(xtx, xty).reshape(-1).cpu().numpy()
        if self.fit_intercept:
            self.coef_ = w[:-1]
            self.intercept_ = float(w[-1])
        else:
            self.coef_ = w
            self.intercept_ = 0.0
        return self

    def predict(self, X) -> np.ndarray:
        dev = self._device()
        Xm = _as_matrix(X, dev)
        w = torch.from_numpy(self.coef_.astype(np.float32)).to(dev)
        out = (Xm @ w) + self.intercept_
        return out.cpu().numpy()

    def score(self, X, y) -> float:
        yp = self.predict(X)
        yt = _as_matrix(y, "cpu").reshape(-1).numpy()
        partial = np.array([
            float(((yt - yp) ** 2).sum()), float(yt.sum()),
            float((yt ** 2).sum()), float(len(yt))])
        parts = comm.allgather_obj(partial)
        tot = np.sum(parts, axis=0)
        ss_res, s, ss, n = tot
        ss_tot = ss - s * s / n
        return 1.0 - ss_res / ss_tot if ss_tot else 0.0


Ridge = LinearRegression


class LogisticRegression:
    """Binary logistic regression, full-batch gradient descent; the X^T r
    gradient GEMM runs on the MFMA path, gradients all-reduced per step."""

    def __init__(self, lr: float = 0.5, max_iter: int = 200,
                 fit_intercept: bool = True, tol: float = 1e-7):
        self.lr = lr
        self.max_iter = max_iter
        self.fit_intercept = fit_intercept
        self.tol = tol
        self.coef_: Optional[np.ndarray] = None
        self.intercept_: float = 0.0

    def fit(self, X, y):
        from .. import config

        dev = torch.device(config.default_device())
        Xm = _as_matrix(X, dev)
        ym = _as_matrix(y, dev).reshape(-1, 1)
        if self.fit_intercept:
            Xm = torch.cat([Xm, torch.ones(Xm.shape[0], 1, dtype=torch.float32,
                                           device=dev)], dim=1)
        n_total = float(sum(comm.allgather_obj(Xm.shape[0])))
        w = torch.zeros(Xm.shape[1], 1, dtype=torch.float32, device=dev)
        prev = None
        for _ in range(self.max_iter):
            z = Xm @ w
            p = torch.sigmoid(z)
            grad = _matmul_at_b(Xm, (p - ym)).to(torch.float64)
            _allreduce_(grad)
            grad = (grad / n_total).to(torch.float32)
            w = w - self.lr * grad
            gn = float(grad.norm().item())
            if prev is not None and abs(prev - gn) < self.tol:
                break
            prev = gn
        wf = w.reshape(-1).cpu().numpy()
        if self.fit_intercept:
            self.coef_, self.intercept_ = wf[:-1], float(wf[-1])
        else:
            self.coef_, self.intercept_ = wf, 0.0
        return self

    def predict_proba(self, X) -> np.ndarray:
        from .. import config

        dev = torch.device(config.default_device())
        Xm = _as_matrix(X, dev)
        w = torch.from_numpy(self.coef_.astype(np.float32)).to(dev)
        z = Xm @ w + self.intercept_
        return torch.sigmoid(z).cpu().numpy()

    def predict(self, X) -> np.ndarray:
        return (self.predict_proba(X) >= 0.5).astype(np.int64)

    def score(self, X, y) -> float:
        """Accuracy, combined across ranks (sklearn surface)."""
        hit = float((self.predict(X) == np.asarray(y)).sum())
        n = float(len(np.asarray(y)))
        if comm.get_world_size() > 1:
            parts = comm.allgather_obj((hit, n))
            hit = sum(p[0] for p in parts)
            n = sum(p[1] for p in parts)
        return hit / max(n, 1.0)


def train_test_split(X, y=None, test_size: float = 0.25, random_state=None):
    """Shard-local split (rows already distributed; reference:
    sklearn_ext train_test_split overloads)."""
    n = len(X)
    rng = np.random.default_rng(random_state)
    perm = rng.permutation(n)
    k = int(round(n * (1 - test_size)))
    tr, te = perm[:k], perm[k:]
    def take(obj, idx):
        if isinstance(obj, (pd.DataFrame, pd.Series)):
            return obj.iloc[idx].reset_index(drop=True)
        return obj[idx]

    if y is None:
        return take(X, tr), take(X, te)
    return take(X, tr), take(X, te), take(y, tr), take(y, te)


def torch_train(model, dataset, loss_fn, optimizer_cls=torch.optim.SGD,
                epochs: int = 1, batch_size: int = 1024, lr: float = 0.01):
    """DDP-style training bridge (reference: bodo/ai/train.py torch_train):
    wraps the model in DistributedDataParallel over the engine's process
    group and iterates the rank-local shard."""
    from .. import config

    dev = torch.device(config.default_device())
    model = model.to(dev)
    if comm.initialized() and comm.get_world_size() > 1:
        from torch.nn.parallel import DistributedDataParallel as DDP

        model = DDP(model)
    opt = optimizer_cls(model.parameters(), lr=lr)
    Xm, ym = dataset
    Xm = _as_matrix(Xm, dev)
    ym = _as_matrix(ym, dev)
    n = Xm.shape[0]
    for _ in range(epochs):
        for s in range(0, n, batch_size):
            xb, yb = Xm[s:s + batch_size], ym[s:s + batch_size]
            opt.zero_grad()
            out = model(xb)
            loss = loss_fn(out, yb)
            loss.backward()
            opt.step()
    return model


class StandardScaler:
    """Distributed standardization: per-feature mean/var from all-reduced
    shard partials (reference: sklearn_ext.py preprocessing overloads)."""

    def __init__(self, with_mean: bool = True, with_std: bool = True):
        self.with_mean = with_mean
        self.with_std = with_std
        self.mean_: Optional[np.ndarray] = None
        self.scale_: Optional[np.ndarray] = None

    def fit(self, X):
        from .. import config

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float64)
        parts = torch.zeros(2 * Xl.shape[1] + 1, dtype=torch.float64,
                            device=Xl.device)
        parts[0] = Xl.shape[0]
        parts[1:Xl.shape[1] + 1] = Xl.sum(dim=0)
        parts[Xl.shape[1] + 1:] = (Xl * Xl).sum(dim=0)
        _allreduce_(parts)
        n = parts[0].item()
        s = parts[1:Xl.shape[1] + 1]
        ss = parts[Xl.shape[1] + 1:]
        mean = s / n
        var = torch.clamp(ss / n - mean * mean, min=0.0)
        self.mean_ = mean.cpu().numpy()
        self.scale_ = np.sqrt(var.cpu().numpy())
        self.scale_[self.scale_ == 0.0] = 1.0
        return self

    def transform(self, X):
        from .. import config

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float64)
        if self.with_mean:
            Xl = Xl - torch.as_tensor(self.mean_, device=Xl.device)
        if self.with_std:
            Xl = Xl / torch.as_tensor(self.scale_, device=Xl.device)
        return Xl.cpu().numpy()

    def fit_transform(self, X):
        return self.fit(X).transform(X)


class KMeans:
    """Distributed Lloyd's iteration: the assignment step is one GEMM
    (X @ C^T, rocBLAS/MFMA on device) per iteration; centroid sums and
    counts are all-reduced over RCCL (reference: sklearn_ext.py KMeans
    fit-with-allreduce pattern)."""

    def __init__(self, n_clusters: int = 8, max_iter: int = 300,
                 tol: float = 1e-6, random_state: Optional[int] = 0):
        self.n_clusters = n_clusters
        self.max_iter = max_iter
        self.tol = tol
        self.random_state = random_state
        self.cluster_centers_: Optional[np.ndarray] = None
        self.inertia_: Optional[float] = None
        self.n_iter_: int = 0

    def _init_centers(self, Xl: torch.Tensor) -> torch.Tensor:
        """Deterministic farthest-point init (greedy k-means++): first
        center is a seeded sample, each next center is the globally
        farthest point from the chosen set (one allgather of (dist, row)
        candidates per center)."""
        k = self.n_clusters
        world = comm.get_world_size() if comm.initialized() else 1
        dev = Xl.device

        def global_pick(val: float, vec: np.ndarray) -> np.ndarray:
            if world == 1:
                return vec
            cands = comm.allgather_obj((float(val), vec))
            return max(cands, key=lambda t: t[0])[1]

        g = torch.Generator().manual_seed(self.random_state or 0)
        first_i = int(torch.randint(max(Xl.shape[0], 1), (1,),
                                    generator=g).item()) % max(Xl.shape[0], 1)
        first = global_pick(Xl.shape[0], Xl[first_i].cpu().numpy())
        centers = [torch.as_tensor(first, device=dev, dtype=Xl.dtype)]
        d2min = ((Xl - centers[0]) ** 2).sum(dim=1)
        for _ in range(1, k):
            j = int(torch.argmax(d2min).item())
            vec = global_pick(d2min[j].item(), Xl[j].cpu().numpy())
            c = torch.as_tensor(vec, device=dev, dtype=Xl.dtype)
            centers.append(c)
            d2min = torch.minimum(d2min, ((Xl - c) ** 2).sum(dim=1))
        return torch.stack(centers)

    def fit(self, X):
        from .. import config

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float32)
        C = self._init_centers(Xl)
        x2 = (Xl * Xl).sum(dim=1, keepdim=True)
        k, d = C.shape
        for it in range(self.max_iter):
            c2 = (C * C).sum(dim=1)
            d2 = x2 - 2.0 * (Xl @ C.t()) + c2
            assign = torch.argmin(d2, dim=1)
            sums = torch.zeros((k, d), dtype=torch.float64, device=dev)
            sums.index_add_(0, assign, Xl.to(torch.float64))
            counts = torch.bincount(assign, minlength=k).to(torch.float64)
            local_inertia = torch.gather(
                d2, 1, assign.view(-1, 1)).clamp_min(0).sum()
            packed = torch.cat([sums.reshape(-1), counts,
                                local_inertia.to(torch.float64).reshape(1)])
            _allreduce_(packed)
            sums = packed[:k * d].reshape(k, d)
            counts = packed[k * d:k * d + k]
            self.inertia_ = float(packed[-1].item())
            newC = torch.where(counts.view(-1, 1) > 0,
                               (sums / counts.clamp(min=1.0).view(-1, 1)),
                               C.to(torch.float64)).to(torch.float32)
            shift = float(((newC - C) ** 2).sum().item())
            C = newC
            self.n_iter_ = it + 1
            if shift <= self.tol:
                break
        self.cluster_centers_ = C.cpu().numpy()
        return self

    def predict(self, X) -> np.ndarray:
        from .. import config

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float32)
        C = torch.as_tensor(self.cluster_centers_, device=dev)
        d2 = (Xl * Xl).sum(dim=1, keepdim=True) - 2.0 * (Xl @ C.t()) \
            + (C * C).sum(dim=1)
        return torch.argmin(d2, dim=1).cpu().numpy()

    def fit_predict(self, X) -> np.ndarray:
        self.fit(X)
        return self.predict(X)


from .gbdt import (GradientBoostingClassifier,  # noqa: E402,F401
                   GradientBoostingRegressor)


class MinMaxScaler:
    """Distributed min-max scaling: per-feature min/max via all-reduce
    (reference: sklearn_ext.py preprocessing overloads)."""

    def __init__(self, feature_range=(0.0, 1.0)):
        self.feature_range = feature_range
        self.data_min_: Optional[np.ndarray] = None
        self.data_max_: Optional[np.ndarray] = None

    def fit(self, X):
        from .. import config
        from ..parallel import comm

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float64)
        lo = Xl.min(dim=0).values if len(Xl) else torch.full(
            (Xl.shape[1],), float("inf"), dtype=torch.float64, device=dev)
        hi = Xl.max(dim=0).values if len(Xl) else torch.full(
            (Xl.shape[1],), float("-inf"), dtype=torch.float64, device=dev)
        if comm.initialized() and comm.get_world_size() > 1:
            import torch.distributed as dist

            dist.all_reduce(lo, op=dist.ReduceOp.MIN)
            dist.all_reduce(hi, op=dist.ReduceOp.MAX)
        self.data_min_ = lo.cpu().numpy()
        self.data_max_ = hi.cpu().numpy()
        return self

    def transform(self, X):
        from .. import config

        dev = torch.device(config.default_device())
        Xl = _as_matrix(X, dev).to(torch.float64).cpu().numpy()
        rng = self.data_max_ - self.data_min_
        rng[rng == 0.0] = 1.0
        a, b = self.feature_range
        return (Xl - self.data_min_) / rng * (b - a) + a

    def fit_transform(self, X):
        return self.fit(X).transform(X)


class LabelEncoder:
    """Distributed label encoding: the class set is the union over shards
    (allgather of locally unique labels)."""

    def __init__(self):
        self.classes_: Optional[np.ndarray] = None

    def fit(self, y):
        from ..parallel import comm

        vals = np.asarray(y)
        uniq = set(pd_unique_list(vals))
        if comm.initialized() and comm.get_world_size() > 1:
            parts = comm.allgather_obj(sorted(uniq, key=str))
            uniq = set()
            for p in parts:
                uniq |= set(p)
        self.classes_ = np.array(sorted(uniq, key=str))
        return self

    def transform(self, y):
        lut = {v: i for i, v in enumerate(self.classes_)}
        return np.array([lut[v] for v in np.asarray(y)], dtype=np.int64)

    def fit_transform(self, y):
        return self.fit(y).transform(y)

    def inverse_transform(self, codes):
        return self.classes_[np.asarray(codes, dtype=np.int64)]


def pd_unique_list(vals):
    import pandas as pd

    return pd.unique(pd.Series(vals).dropna()).tolist()


def train_test_split(X, y=None, test_size=0.25, random_state=None,
                     shuffle=True):
    """Shard-local split with a shared seed (each rank splits its block;
    reference: sklearn_ext.py train_test_split overload)."""
    n = len(X)
    idx = np.arange(n)
    if shuffle:
        rng = np.random.default_rng(random_state)
        rng.shuffle(idx)
    n_test = int(round(n * test_size)) if test_size < 1 else int(test_size)
    test_i, train_i = idx[:n_test], idx[n_test:]

    def take(a, i):
        if hasattr(a, "iloc"):
            return a.iloc[i]
        return np.asarray(a)[i]

    if y is None:
        return take(X, train_i), take(X, test_i)
    return take(X, train_i), take(X, test_i), take(y, train_i), take(y, test_i)


# ---------------------------------------------------------------------------
# metrics (distributed: partials all-reduced)
# ---------------------------------------------------------------------------

def _metric_partials(vals):
    from .. import config

    dev = torch.device(config.default_device())
    t = torch.tensor(vals, dtype=torch.float64, device=dev)
    _allreduce_(t)
    return t.cpu().numpy()


def accuracy_score(y_true, y_pred) -> float:
    yt, yp = np.asarray(y_true), np.asarray(y_pred)
    p = _metric_partials([float((yt == yp).sum()), float(len(yt))])
    return p[0] / p[1] if p[1] else 0.0


def mean_squared_error(y_true, y_pred) -> float:
    yt = np.asarray(y_true, dtype=np.float64)
    yp = np.asarray(y_pred, dtype=np.float64)
    p = _metric_partials([float(((yt - yp) ** 2).sum()), float(len(yt))])
    return p[0] / p[1] if p[1] else 0.0


def mean_absolute_error(y_true, y_pred) -> float:
    yt = np.asarray(y_true, dtype=np.float64)
    yp = np.asarray(y_pred, dtype=np.float64)
    p = _metric_partials([float(np.abs(yt - yp).sum()), float(len(yt))])
    return p[0] / p[1] if p[1] else 0.0


def r2_score(y_true, y_pred) -> float:
    yt = np.asarray(y_true, dtype=np.float64)
    yp = np.asarray(y_pred, dtype=np.float64)
    p = _metric_partials([
        float(((yt - yp) ** 2).sum()), float(yt.sum()),
        float((yt * yt).sum()), float(len(yt))])
    sse, s, ss, n = p
    if n == 0:
        return 0.0
    sst = ss - s * s / n
    return 1.0 - sse / sst if sst else 0.0
