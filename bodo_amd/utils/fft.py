"""FFT over distributed arrays (reference role: bodo/libs/_fft.cpp —
FFTW+MPI).  MI355X design: transforms run on-device through torch.fft
(rocFFT underneath); block-distributed inputs replicate for the transform
(a transform mixes every element) and the result re-scatters to blocks.
Suitable up to HBM-sized signals; a slab-decomposed multi-GPU pipeline is
the extension point."""

from __future__ import annotations

import numpy as np
import torch

from ..compiler.distarray import DistArray, _block_bounds
from ..parallel import comm


def _gather_tensor(a):
    if isinstance(a, DistArray):
        if comm.get_world_size() > 1:
            parts = comm.allgather_obj(a.t.cpu())
            return torch.cat(parts).to(a.t.device), a.total, True
        return a.t, a.total, True
    if isinstance(a, np.ndarray):
        return torch.from_numpy(a), len(a), False
    return a, int(a.numel()), False


def _rescatter(full: torch.Tensor, dist: bool):
    if not dist:
        return full.cpu().numpy()
    w, r = comm.get_world_size(), comm.get_rank()
    n = int(full.numel())
    s, e = _block_bounds(n, w, r)
    return DistArray(full[s:e].clone(), n)


def fft(a):
    t, n, dist = _gather_tensor(a)
    return _rescatter(torch.fft.fft(t), dist)


def ifft(a):
    t, n, dist = _gather_tensor(a)
    return _rescatter(torch.fft.ifft(t), dist)


def rfft(a):
    t, n, dist = _gather_tensor(a)
    return _rescatter(torch.fft.rfft(t), dist)


def irfft(a, n=None):
    t, _, dist = _gather_tensor(a)
    return _rescatter(torch.fft.irfft(t, n=n), dist)


def fftfreq(n, d=1.0):
    return torch.fft.fftfreq(n, d=d).numpy()
