"""@bodo_amd.jit (reference: bodo/decorators.py:338).

Round-1 semantics: the decorated function runs SPMD with ``bodo_amd.pandas``
substituted for pandas inside (dataframe-library mode, the reference's
df-lib check_func mode bodo/tests/utils.py:236-243).  The Numba-typed
distributed-IR pipeline with HIP lowering is the upgrade path (SURVEY §7
step 7)."""

from __future__ import annotations

import functools


def jit(fn=None, **options):
    if fn is None:
        return lambda f: jit(f, **options)

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        return fn(*args, **kwargs)

    wrapper._is_bodo_jit = True
    wrapper.py_func = fn
    return wrapper
