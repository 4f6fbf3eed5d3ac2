"""Bit-faithful host simulators of the csrc/parquet.hip kernels with hard
bounds assertions.  Used by CPU tests to validate the device decode logic
(index arithmetic, parse state machines) without a GPU; any out-of-bounds
access that would be a hardware fault on the MI355X raises here."""

import numpy as np
import torch

PQF_HAS_DEF = 1
PQF_SNAPPY = 2


class Buf:
    """Bounds-checked byte view."""

    def __init__(self, arr: np.ndarray, lo=0, hi=None, slack=0):
        self.a = arr
        self.lo = lo
        self.hi = len(arr) if hi is None else hi
        self.slack = slack  # allowed overread (scratch pad)

    def __getitem__(self, i):
        assert self.lo + i < self.hi + self.slack, \
            f"OOB read at +{i} (len {self.hi - self.lo}, slack {self.slack})"
        assert i >= 0, f"negative index {i}"
        j = self.lo + i
        return int(self.a[j]) if j < len(self.a) else 0

    def set(self, i, v):
        assert 0 <= i and self.lo + i < self.hi, f"OOB write at +{i}"
        self.a[self.lo + i] = v


def sim_decompress(src: np.ndarray, metas, scratch: np.ndarray):
    for pm in metas:
        inb = Buf(src, int(pm["src_off"]), int(pm["src_off"]) + int(pm["src_len"]))
        out = Buf(scratch, int(pm["dst_off"]), int(pm["dst_off"]) + int(pm["dst_len"]))
        if not (int(pm["flags"]) & PQF_SNAPPY):
            for i in range(int(pm["dst_len"])):
                out.set(i, inb[i])
            continue
        ip, opos = 0, 0
        src_len, dst_len = int(pm["src_len"]), int(pm["dst_len"])
        while inb[ip] & 0x80:
            ip += 1
        ip += 1
        while ip < src_len and opos < dst_len:
            tag = inb[ip]
            k = tag & 3
            if k == 0:
                ln = (tag >> 2) + 1
                hl = 1
                if ln > 60:
                    nb = ln - 60
                    ln = 0
                    for i in range(nb):
                        ln |= inb[ip + 1 + i] << (8 * i)
                    ln += 1
                    hl = 1 + nb
                srcoff = ip + hl
                ip += hl + ln
                assert ln > 0 and opos + ln <= dst_len, "literal overrun"
                for i in range(ln):
                    out.set(opos + i, inb[srcoff + i])
            else:
                if k == 1:
                    ln = ((tag >> 2) & 7) + 4
                    off = ((tag >> 5) << 8) | inb[ip + 1]
                    hl = 2
                elif k == 2:
                    ln = (tag >> 2) + 1
                    off = inb[ip + 1] | (inb[ip + 2] << 8)
                    hl = 3
                else:
                    ln = (tag >> 2) + 1
                    off = (inb[ip + 1] | (inb[ip + 2] << 8)
                           | (inb[ip + 3] << 16) | (inb[ip + 4] << 24))
                    hl = 5
                ip += hl
                assert 0 < off <= opos, f"bad copy offset {off} at opos {opos}"
                assert opos + ln <= dst_len, "copy overrun"
                for i in range(ln):
                    out.set(opos + i, out[opos - off + (i % off if off < ln else i)])
            opos += ln
        assert opos == dst_len, f"decompress short: {opos} != {dst_len}"


def _varint(buf: Buf, pos):
    out, shift = 0, 0
    while True:
        v = buf[pos]
        pos += 1
        out |= (v & 0x7F) << shift
        if not (v & 0x80):
            return out, pos
        shift += 7
        assert shift < 64, "runaway varint"


def sim_def_levels(scratch, metas, bitwidth, max_def, total_nv, slack=16):
    mask = np.zeros(total_nv, dtype=np.uint8)
    n_valid = np.zeros(len(metas), dtype=np.int32)
    vdo = np.zeros(len(metas), dtype=np.int32)
    for p, pm in enumerate(metas):
        base = Buf(scratch, int(pm["dst_off"]),
                   int(pm["dst_off"]) + int(pm["dst_len"]), slack)
        nv = int(pm["nv"])
        mv = Buf(mask, int(pm["val_off"]), int(pm["val_off"]) + nv)
        if not (int(pm["flags"]) & PQF_HAS_DEF):
            for i in range(nv):
                mv.set(i, 1)
            n_valid[p] = nv
            continue
        lvl_len = base[0] | (base[1] << 8) | (base[2] << 16) | (base[3] << 24)
        vdo[p] = 4 + lvl_len
        lv = Buf(scratch, int(pm["dst_off"]) + 4,
                 int(pm["dst_off"]) + int(pm["dst_len"]), slack)
        pos, out, valid = 0, 0, 0
        wb = (bitwidth + 7) // 8
        while out < nv and pos < lvl_len:
            h, pos = _varint(lv, pos)
            if h & 1:
                groups = h >> 1
                count = min(groups * 8, nv - out)
                bitoff = pos * 8
                for i in range(count):
                    bp = bitoff + i * bitwidth
                    w = lv[bp >> 3] | (lv[(bp >> 3) + 1] << 8)
                    v = (w >> (bp & 7)) & ((1 << bitwidth) - 1)
                    ok = 1 if v == max_def else 0
                    mv.set(out + i, ok)
                    valid += ok
                pos += groups * bitwidth
            else:
                count = min(h >> 1, nv - out)
                val = 0
                for i in range(wb):
                    val |= lv[pos + i] << (8 * i)
                pos += wb
                ok = 1 if val == max_def else 0
                for i in range(count):
                    mv.set(out + i, ok)
                valid += ok * count
            out += count
        assert out == nv, f"def levels short: {out} != {nv}"
        n_valid[p] = valid
    return mask, n_valid, vdo


def sim_expand_codes(scratch, metas, vdo, dense_off, n_valid, dense_total,
                     slack=16):
    out = np.zeros(dense_total, dtype=np.int32)
    for p, pm in enumerate(metas):
        start = int(pm["dst_off"]) + (int(vdo[p]) if vdo is not None else 0)
        base = Buf(scratch, start,
                   int(pm["dst_off"]) + int(pm["dst_len"]), slack)
        nval = int(n_valid[p]) if n_valid is not None else int(pm["nv"])
        o = Buf(out, int(dense_off[p]), int(dense_off[p]) + nval)
        bitwidth = base[0]
        d = Buf(scratch, start + 1, int(pm["dst_off"]) + int(pm["dst_len"]),
                slack)
        if bitwidth == 0:
            for i in range(nval):
                o.set(i, 0)
            continue
        assert 0 < bitwidth <= 24, f"bitwidth {bitwidth}"
        wb = (bitwidth + 7) // 8
        pos, outp = 0, 0
        dlen = int(pm["dst_len"]) - (start + 1 - int(pm["dst_off"]))
        while outp < nval and pos < dlen:
            h, pos = _varint(d, pos)
            if h & 1:
                groups = h >> 1
                count = min(groups * 8, nval - outp)
                bitoff = pos * 8
                for i in range(count):
                    bp = bitoff + i * bitwidth
                    byte = bp >> 3
                    w = (d[byte] | (d[byte + 1] << 8) | (d[byte + 2] << 16)
                         | (d[byte + 3] << 24))
                    o.set(outp + i, (w >> (bp & 7)) & ((1 << bitwidth) - 1))
                pos += groups * bitwidth
            else:
                count = min(h >> 1, nval - outp)
                val = 0
                for i in range(wb):
                    val |= d[pos + i] << (8 * i)
                pos += wb
                for i in range(count):
                    o.set(outp + i, val)
            outp += count
        assert outp == nval, f"codes short: {outp} != {nval}"
    return out


def sim_copy_fixed(scratch, metas, vdo, dense_off, n_valid, esize,
                   dense_total):
    out = np.zeros(dense_total * esize, dtype=np.uint8)
    for p, pm in enumerate(metas):
        start = int(pm["dst_off"]) + (int(vdo[p]) if vdo is not None else 0)
        nval = int(n_valid[p]) if n_valid is not None else int(pm["nv"])
        nbytes = nval * esize
        assert start + nbytes <= int(pm["dst_off"]) + int(pm["dst_len"]), \
            f"fixed copy source overrun page {p}"
        do = int(dense_off[p]) * esize
        out[do:do + nbytes] = scratch[start:start + nbytes]
    return out


def sim_byte_array(scratch, metas, vdo, dense_off, n_valid, dense_total):
    lengths = np.zeros(dense_total, dtype=np.int32)
    src_abs = np.zeros(dense_total, dtype=np.int64)
    for p, pm in enumerate(metas):
        voff = int(pm["dst_off"]) + (int(vdo[p]) if vdo is not None else 0)
        s = Buf(scratch, voff, int(pm["dst_off"]) + int(pm["dst_len"]))
        nval = int(n_valid[p]) if n_valid is not None else int(pm["nv"])
        pos = 0
        for i in range(nval):
            ln = s[pos] | (s[pos + 1] << 8) | (s[pos + 2] << 16) | (s[pos + 3] << 24)
            lengths[int(dense_off[p]) + i] = ln
            src_abs[int(dense_off[p]) + i] = voff + pos + 4
            pos += 4 + ln
        assert pos <= int(pm["dst_len"]) - (voff - int(pm["dst_off"])), \
            f"byte array walk overran page {p}"
    return lengths, src_abs
