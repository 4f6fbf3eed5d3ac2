"""Drop-in alias for the reference package name: ``import bodo.pandas as pd``
and ``@bodo.jit`` resolve to the MI355X-native bodo_amd implementation."""

from bodo_amd import (  # noqa: F401
    __version__, allgatherv, barrier, gatherv, get_gpu_ranks, get_rank,
    get_size, jit, random_shuffle, rebalance, scatterv, wrap_python,
)
from bodo_amd import config  # noqa: F401
