"""Lazy scalar: the result of a reduction stays a plan until first use
(reference: BodoScalar, bodo/pandas/scalar.py:14 — a `.sum()` in the middle
of a pipeline must not force a collection round; the value materializes on
the first arithmetic/comparison/format access)."""

from __future__ import annotations

_UNSET = object()


class BodoScalar:
    __slots__ = ("_plan", "_col", "_value")

    def __init__(self, plan, col: str = "r"):
        self._plan = plan
        self._col = col
        self._value = _UNSET

    @property
    def value(self):
        if self._value is _UNSET:
            from ..engine import api

            df = api.collect(self._plan)
            self._value = df[self._col].iloc[0] if len(df) else None
            self._plan = None
        return self._value

    def item(self):
        return self.value

    # ---- conversions / display ----
    def __float__(self):
        return float(self.value)

    def __int__(self):
        return int(self.value)

    def __bool__(self):
        return bool(self.value)

    def __index__(self):
        return int(self.value)

    def __repr__(self):
        return repr(self.value)

    def __str__(self):
        return str(self.value)

    def __format__(self, spec):
        return format(self.value, spec)

    def __hash__(self):
        return hash(self.value)

    def __array__(self, dtype=None, copy=None):
        import numpy as np

        return np.asarray(self.value, dtype=dtype)

    def __round__(self, n=None):
        return round(self.value, n) if n is not None else round(self.value)

    def __abs__(self):
        return abs(self.value)

    def __neg__(self):
        return -self.value

    # ---- arithmetic (materializing; results are plain scalars) ----
    def __add__(self, o):
        return self.value + _unwrap(o)

    def __radd__(self, o):
        return _unwrap(o) + self.value

    def __sub__(self, o):
        return self.value - _unwrap(o)

    def __rsub__(self, o):
        return _unwrap(o) - self.value

    def __mul__(self, o):
        return self.value * _unwrap(o)

    def __rmul__(self, o):
        return _unwrap(o) * self.value

    def __truediv__(self, o):
        return self.value / _unwrap(o)

    def __rtruediv__(self, o):
        return _unwrap(o) / self.value

    def __floordiv__(self, o):
        return self.value // _unwrap(o)

    def __mod__(self, o):
        return self.value % _unwrap(o)

    def __pow__(self, o):
        return self.value ** _unwrap(o)

    # ---- comparisons ----
    def __eq__(self, o):
        return self.value == _unwrap(o)

    def __ne__(self, o):
        return self.value != _unwrap(o)

    def __lt__(self, o):
        return self.value < _unwrap(o)

    def __le__(self, o):
        return self.value <= _unwrap(o)

    def __gt__(self, o):
        return self.value > _unwrap(o)

    def __ge__(self, o):
        return self.value >= _unwrap(o)


def _unwrap(v):
    return v.value if isinstance(v, BodoScalar) else v
