from bodo_amd.pandas import *  # noqa: F401,F403
from bodo_amd.pandas import (  # noqa: F401
    BodoDataFrame, BodoSeries, DataFrame, NamedAgg, Series, Timestamp,
    concat, from_pandas, merge, read_csv, read_json, read_parquet,
    to_datetime,
)


def __getattr__(name):
    import bodo_amd.pandas as _m

    return getattr(_m, name)
