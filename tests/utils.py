"""Test utilities: check_query is the analog of the reference's check_func
(bodo/tests/utils.py:157) — run a query through bodo_amd.pandas and compare
against plain pandas."""

from __future__ import annotations

import numpy as np
import pandas as pd

import bodo_amd.pandas as bpd


def check_query(fn, inputs: dict, sort_by=None, check_dtype=False,
                reset_index=True, atol=1e-8, rtol=1e-6):
    """fn(pdmod, **inputs_as_frames) -> DataFrame-like.

    Runs fn twice: once with real pandas (inputs passed through) and once
    with bodo_amd.pandas (inputs wrapped via from_pandas); asserts equality.
    """
    exp = fn(pd, **{k: v.copy() for k, v in inputs.items()})
    got = fn(bpd, **{k: bpd.from_pandas(v) for k, v in inputs.items()})
    if hasattr(got, "to_pandas"):
        got = got.to_pandas()
    if isinstance(exp, pd.Series):
        exp = exp.to_frame()
    if isinstance(got, pd.Series):
        got = got.to_frame()
    if isinstance(exp, pd.DataFrame):
        if sort_by:
            exp = exp.sort_values(sort_by)
            got = got.sort_values(sort_by)
        if reset_index:
            exp = exp.reset_index(drop=True)
            got = got.reset_index(drop=True)
        got = _normalize(got)
        exp = _normalize(exp)
        pd.testing.assert_frame_equal(got, exp, check_dtype=check_dtype,
                                      atol=atol, rtol=rtol)
    else:
        if isinstance(exp, float) and isinstance(got, float):
            assert abs(exp - got) <= atol + rtol * abs(exp), (got, exp)
        else:
            assert got == exp, (got, exp)
    return got


def _normalize(df: pd.DataFrame) -> pd.DataFrame:
    out = df.copy()
    for c in out.columns:
        if isinstance(out[c].dtype, pd.CategoricalDtype):
            out[c] = out[c].astype(str)
        elif out[c].dtype == object:
            out[c] = out[c].astype(str).where(~out[c].isna(), np.nan)
    return out
