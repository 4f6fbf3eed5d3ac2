"""numpy facade bound as `np` inside @bodo_amd.jit bodies: array CREATION
functions return block-distributed DistArrays (>= threshold elements);
everything else forwards to real numpy, whose ufuncs/function protocol then
dispatch back onto DistArray shards.  Reference role: the untyped/typing
passes recognizing np.* allocation calls (bodo/transforms/untyped_pass.py)
— here the module boundary is the recognition point."""

from __future__ import annotations

import numpy as _np

from .distarray import DistArray

# arrays smaller than this stay replicated numpy (REP inference for small
# constants, reference: Distribution.REP)
DIST_THRESHOLD = 1024


def _n(shape):
    if isinstance(shape, (tuple, list)):
        if len(shape) != 1:
            return None
        shape = shape[0]
    return int(shape)


def arange(*args, **kwargs):
    if len(args) == 1 and not kwargs:
        n = _n(args[0])
        if n is not None and n >= DIST_THRESHOLD:
            return DistArray.arange(n)
    return _np.arange(*args, **kwargs)


def zeros(shape, dtype=float):
    n = _n(shape)
    if n is not None and n >= DIST_THRESHOLD:
        return DistArray.full(n, 0, dtype)
    return _np.zeros(shape, dtype)


def ones(shape, dtype=float):
    n = _n(shape)
    if n is not None and n >= DIST_THRESHOLD:
        return DistArray.full(n, 1, dtype)
    return _np.ones(shape, dtype)


def empty(shape, dtype=float):
    return zeros(shape, dtype)


def full(shape, fill, dtype=None):
    n = _n(shape)
    if n is not None and n >= DIST_THRESHOLD:
        return DistArray.full(n, fill, dtype)
    return _np.full(shape, fill, dtype)


class _RandomShim:
    def __getattr__(self, name):
        return getattr(_np.random, name)

    @staticmethod
    def _dist(n, seed=None):
        return DistArray.random(n, seed)

    def ranf(self, n=None):
        if n is not None and _n(n) and _n(n) >= DIST_THRESHOLD:
            return self._dist(_n(n))
        return _np.random.ranf(n)

    def rand(self, *shape):
        if len(shape) == 1 and shape[0] >= DIST_THRESHOLD:
            return self._dist(shape[0])
        return _np.random.rand(*shape)

    def random(self, n=None):
        if n is not None and _n(n) and _n(n) >= DIST_THRESHOLD:
            return self._dist(_n(n))
        return _np.random.random(n)


random = _RandomShim()


def __getattr__(name):
    return getattr(_np, name)
