"""@bodo_amd.jit (reference: bodo/decorators.py:338).

Round-1 semantics: the decorated function runs SPMD with ``bodo_amd.pandas``
substituted for pandas inside (dataframe-library mode, the reference's
df-lib check_func mode bodo/tests/utils.py:236-243).  The Numba-typed
distributed-IR pipeline with HIP lowering is the upgrade path (SURVEY §7
step 7)."""

from __future__ import annotations

import functools


_KNOWN_OPTIONS = {
    "distributed", "replicated", "spawn", "cache", "returns_maybe_distributed",
    "args_maybe_distributed", "all_args_distributed_block",
    "all_args_distributed_varlength", "all_returns_distributed",
    "distributed_diagnostics", "pivots", "inline", "parallel",
}


def _df_lib_call(fn, args, kwargs):
    """DataFrame-library execution of a jitted function (reference:
    check_func df-lib mode, bodo/tests/utils.py:236-243): globals bound to
    the real pandas module rebind to bodo_amd.pandas for the call, and
    pandas DataFrame arguments become lazy distributed frames, so the
    function body plans through the engine instead of eager pandas."""
    import pandas as real_pd

    import bodo_amd.pandas as bpd

    g = getattr(fn, "__globals__", None)
    replaced = []
    if isinstance(g, dict):
        replaced = [k for k, v in g.items() if v is real_pd]
    conv = [bpd.from_pandas(a) if isinstance(a, real_pd.DataFrame) else a
            for a in args]
    try:
        for k in replaced:
            g[k] = bpd
        return fn(*conv, **kwargs)
    finally:
        for k in replaced:
            g[k] = real_pd


def jit(fn=None, **options):
    """Accepts the reference's jit option surface (decorators.py:183-230);
    the df-library execution model makes most of them no-ops here, while
    `cache` keeps compiled HIP UDF modules across calls (hip_udf caches by
    source) and `spawn` routes execution through spawn-mode workers when
    BODO_NUM_WORKERS is set."""
    unknown = set(options) - _KNOWN_OPTIONS
    if unknown:
        import warnings

        warnings.warn(f"bodo_amd.jit: ignoring unknown options {unknown}")
    if fn is None:
        return lambda f: jit(f, **options)

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        from ..parallel import spawn

        if spawn.active():
            import pandas as pd

            # functions whose args are plain host data are shipped to the
            # workers (reference: SpawnDispatcher, spawner.py:1029)
            if all(not hasattr(a, "_lazy_plan") for a in args):
                sp = spawn.get_spawner()
                reps = sp.exec_func(fn, list(args), kwargs)
                r0 = reps[0]
                if r0.get("kind") == "frame":
                    from ..pandas.frame import BodoDataFrame
                    from ..plan import nodes as pn

                    out = BodoDataFrame(
                        pn.PandasScan(r0["res_id"], tuple(r0["names"]),
                                      distributed=True), list(r0["names"]))
                    object.__setattr__(out, "_remote", spawn.RemoteResult(
                        r0["res_id"], r0["names"], r0["length"]))
                    return out
                return r0.get("value")
        return _df_lib_call(fn, args, kwargs)

    wrapper._is_bodo_jit = True
    wrapper.py_func = fn
    wrapper.targetoptions = options
    return wrapper
