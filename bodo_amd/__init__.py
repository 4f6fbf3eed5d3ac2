"""bodo_amd: an MI355X-native distributed DataFrame/SQL engine with the
capabilities of bodo-ai/Bodo, built from scratch for CDNA4 (gfx950):

* ``bodo_amd.pandas`` — lazy drop-in pandas API (logical plans, optimizer,
  columnar HBM-resident executor with hand-written HIP kernels)
* ``@bodo_amd.jit`` — function decorator executing over distributed frames
* one process per GPU, ``torch.distributed`` over RCCL/xGMI for the data
  plane (hash shuffles, broadcasts), gloo for CPU test runs

Reference (behavioral spec only): bodo-ai/Bodo @ /root/reference.
"""

from __future__ import annotations

import os

__version__ = "0.1.0"

from . import config  # noqa: F401

# SPMD mode: auto-init the process group when launched under torchrun
# (reference analog: bodo spawn-mode workers init MPI at import)
if "RANK" in os.environ and "MASTER_ADDR" in os.environ:
    from .parallel import comm as _comm

    _comm.init_from_env()


def jit(fn=None, **options):
    from .jit.decorator import jit as _jit

    return _jit(fn, **options)


def get_rank() -> int:
    from .parallel import comm

    return comm.get_rank()


def get_size() -> int:
    from .parallel import comm

    return comm.get_world_size()


def barrier():
    from .parallel import comm

    comm.barrier()
