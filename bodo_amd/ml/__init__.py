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
