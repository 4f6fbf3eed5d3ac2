"""Cardinality / quantile sketches as tensor ops (reference role:
bodo/libs/_theta_sketches.cpp + vendored hyperloglog.hpp + _bodo_tdigest.cpp
— redesigned: HLL registers are a dense device tensor updated with one
scatter-max, merged across ranks by an elementwise MAX all-reduce over RCCL;
the quantile sketch is a bounded uniform sample)."""

from __future__ import annotations

import math

import torch

from ..core.column import Column

HLL_P = 14  # 2^14 registers = 16 KiB, ~0.8% relative error


def hll_registers(cols, p: int = HLL_P) -> torch.Tensor:
    """Build HLL registers (int64 tensor of 2^p entries, values 0..64) from
    the row hashes of `cols` on their device."""
    from .. import ops

    h = ops.hash_columns(cols)  # int64 row hashes, device-resident
    m = 1 << p
    bucket = (h & (m - 1)).long()
    rest = (h >> p) & ((1 << (64 - p)) - 1)  # logical shift of the top bits
    # rank = leading-zero count within the (64-p)-bit field + 1; computed as
    # (64-p) - floor(log2(rest)) for rest>0 via the float exponent
    nz = rest != 0
    # bit_length via conversion through float64 is exact for < 2^53; the
    # top (64-p)=50 bits fit
    fl = rest.clamp(min=1).double()
    bl = torch.floor(torch.log2(fl)).long() + 1  # bit length
    rank = torch.where(nz, (64 - p) - bl + 1,
                       torch.full_like(bl, 64 - p + 1))
    regs = torch.zeros(m, dtype=torch.int64, device=h.device)
    regs.scatter_reduce_(0, bucket, rank, reduce="amax")
    return regs


def hll_estimate(regs: torch.Tensor) -> float:
    m = regs.numel()
    alpha = 0.7213 / (1 + 1.079 / m)
    inv = torch.pow(2.0, -regs.double())
    e = alpha * m * m / float(inv.sum().item())
    zeros = int((regs == 0).sum().item())
    if e <= 2.5 * m and zeros:
        e = m * math.log(m / zeros)  # linear counting, small range
    return float(e)


def hll_merge_(regs: torch.Tensor) -> torch.Tensor:
    """Elementwise MAX across ranks (RCCL all-reduce), in place."""
    from ..parallel import comm

    comm.allreduce_max_(regs)
    return regs


def approx_nunique(cols, distributed: bool = True) -> float:
    regs = hll_registers(cols if isinstance(cols, list) else [cols])
    if distributed:
        hll_merge_(regs)
    return hll_estimate(regs)


# ---------------------------------------------------------------------
# quantile sketch: bounded uniform sample (t-digest role for APPROX_PERCENTILE)
# ---------------------------------------------------------------------

SAMPLE_CAP = 1 << 17


def quantile_sample(col: Column, cap: int = SAMPLE_CAP) -> torch.Tensor:
    data = col.data
    if col.mask is not None:
        data = data[col.mask]
    if col.dtype.is_float:
        data = data[~torch.isnan(data)]
    n = data.numel()
    if n > cap:
        g = torch.Generator(device="cpu")
        g.manual_seed(0xD1)
        pos = torch.randint(0, n, (cap,), generator=g).to(data.device)
        data = data[pos]
    return data


def approx_percentile(col: Column, q: float, distributed: bool = True) -> float:
    from ..parallel import comm

    sample = quantile_sample(col).double()
    if distributed and comm.get_world_size() > 1:
        parts = comm.allgather_obj(sample.cpu())
        sample = torch.cat([p for p in parts]).to(sample.device)
    if sample.numel() == 0:
        return float("nan")
    return float(torch.quantile(sample, q).item())
