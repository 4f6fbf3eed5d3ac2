"""Env-var driven configuration (reference: bodo/__init__.py:104-237 flag
system).  One module, read at import, overridable for tests."""

from __future__ import annotations

import os

import torch


def _env_int(name, default):
    try:
        return int(os.environ.get(name, default))
    except ValueError:
        return default


def _env_bool(name, default):
    v = os.environ.get(name)
    if v is None:
        return default
    return v.lower() in ("1", "true", "yes", "on")


#: force device: "cuda", "cpu", or "" (auto)
DEVICE = os.environ.get("BODO_AMD_DEVICE", "")

#: broadcast-join threshold in bytes (reference: BODO_BCAST_JOIN_THRESHOLD)
BCAST_JOIN_THRESHOLD = _env_int("BODO_AMD_BCAST_JOIN_THRESHOLD", 256 << 20)

#: morsel rows for the streaming executor (reference default 32768; sized up
#: for 288 GB HBM per GPU)
STREAM_BATCH_SIZE = _env_int("BODO_AMD_BATCH_SIZE", 4 << 20)

#: require the native HIP extension when running on GPU (fail loudly)
REQUIRE_NATIVE = _env_bool("BODO_AMD_REQUIRE_NATIVE", True)

#: number of workers for spawn mode (reference: BODO_NUM_WORKERS)
NUM_WORKERS = _env_int("BODO_NUM_WORKERS", 0)

#: fall back to real pandas for unimplemented API (reference:
#: BODO_PANDAS_FALLBACK bodo/pandas/__init__.py)
PANDAS_FALLBACK = _env_bool("BODO_AMD_PANDAS_FALLBACK", True)

#: verbosity for user logging
VERBOSE = _env_int("BODO_AMD_VERBOSE", 0)

#: streaming execution: "0" off, "1" force, "auto" = when scan bytes exceed
#: STREAM_THRESHOLD_BYTES (reference: streaming operator states with 288 GB
#: HBM-sized morsels)
STREAMING = os.environ.get("BODO_AMD_STREAMING", "auto")

#: fuse projection/filter expressions into one hipRTC kernel on GPU
FUSE_EXPR = _env_bool("BODO_AMD_FUSE_EXPR", True)

#: auto-streaming threshold (bytes of scanned files per query)
STREAM_THRESHOLD_BYTES = _env_int("BODO_AMD_STREAM_THRESHOLD",
                                  100 * 1024**3)

#: on-disk cache for hipRTC-compiled UDF/fused-expression code objects
#: (the reference's @bodo.jit(cache=True) durable-compile analog,
#: bodo/tests/caching_tests); "" disables
KERNEL_CACHE_DIR = os.environ.get(
    "BODO_AMD_KERNEL_CACHE",
    os.path.join(os.path.expanduser("~"), ".cache", "bodo_amd_kernels"))


def default_device() -> str:
    if DEVICE:
        return DEVICE
    return "cuda" if torch.cuda.is_available() else "cpu"

#: out-of-core budget override in bytes (0 = auto: half of free HBM on GPU,
#: unlimited on CPU).  See engine/ooc.py — partition-splitting hash operators
#: with host-DRAM staging when a local groupby/join exceeds the budget.
OOC_BYTES = _env_int("BODO_AMD_OOC_BYTES", 0)

#: synchronize the device after every operator and surface async HIP errors
#: at the operator that caused them (reference analog: DEBUG_PIPELINE /
#: compute-sanitizer harness)
DEBUG_SYNC = _env_bool("BODO_AMD_DEBUG_SYNC", False)
