"""Synthetic TPC-H data generator (no network: dbgen-shaped, not dbgen).

Counter-based: every value is a pure function of the global row index, so any
rank can generate exactly its shard at any world size and results agree with
a single-rank run.  Value domains cover every predicate the 22 queries test
(brands, types, containers, nations, segments, comment trigger phrases).

Scale: sf=1 gives the standard row counts (lineitem 6M, orders 1.5M, ...).
"""

from __future__ import annotations

import numpy as np
import pandas as pd

U64 = np.uint64
_M1 = U64(0xff51afd7ed558ccd)
_M2 = U64(0xc4ceb9fe1a85ec53)


def _mix(x: np.ndarray, salt: int) -> np.ndarray:
    with np.errstate(over="ignore"):
        x = x.astype(np.uint64) ^ U64(
            (salt * 0x9E3779B97F4A7C15 + 0x1234567) & 0xFFFFFFFFFFFFFFFF)
        x ^= x >> U64(33)
        x *= _M1
        x ^= x >> U64(33)
        x *= _M2
        x ^= x >> U64(33)
    return x


def _uni(idx, salt, lo, hi):
    """uniform integer in [lo, hi]"""
    return (lo + (_mix(idx, salt) % U64(hi - lo + 1)).astype(np.int64))


def _unif(idx, salt, lo, hi):
    """uniform float in [lo, hi)"""
    u = _mix(idx, salt).astype(np.float64) / 2**64
    return lo + u * (hi - lo)


def _pick(idx, salt, values):
    """Categorical (codes + small category list): no per-row python objects,
    scales to SF100 row counts."""
    codes = _uni(idx, salt, 0, len(values) - 1).astype(np.int32)
    return pd.Categorical.from_codes(codes, categories=list(values))


NATIONS = [
    ("ALGERIA", 0), ("ARGENTINA", 1), ("BRAZIL", 1), ("CANADA", 1),
    ("EGYPT", 4), ("ETHIOPIA", 0), ("FRANCE", 3), ("GERMANY", 3),
    ("INDIA", 2), ("INDONESIA", 2), ("IRAN", 4), ("IRAQ", 4), ("JAPAN", 2),
    ("JORDAN", 4), ("KENYA", 0), ("MOROCCO", 0), ("MOZAMBIQUE", 0),
    ("PERU", 1), ("CHINA", 2), ("ROMANIA", 3), ("SAUDI ARABIA", 4),
    ("VIETNAM", 2), ("RUSSIA", 3), ("UNITED KINGDOM", 3),
    ("UNITED STATES", 1),
]
REGIONS = ["AFRICA", "AMERICA", "ASIA", "EUROPE", "MIDDLE EAST"]
SEGMENTS = ["AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD"]
PRIORITIES = ["1-URGENT", "2-HIGH", "3-MEDIUM", "4-NOT SPECIFIED", "5-LOW"]
SHIPMODES = ["REG AIR", "AIR", "RAIL", "SHIP", "TRUCK", "MAIL", "FOB",
             "AIR REG"]
SHIPINSTRUCT = ["DELIVER IN PERSON", "COLLECT COD", "NONE",
                "TAKE BACK RETURN"]
TYPE1 = ["STANDARD", "SMALL", "MEDIUM", "LARGE", "ECONOMY", "PROMO"]
TYPE2 = ["ANODIZED", "BURNISHED", "PLATED", "POLISHED", "BRUSHED"]
TYPE3 = ["TIN", "NICKEL", "BRASS", "STEEL", "COPPER"]
CONT1 = ["SM", "MED", "LG", "JUMBO", "WRAP"]
CONT2 = ["CASE", "BOX", "PACK", "PKG", "BAG", "JAR", "DRUM", "CAN"]
COLORS = ["almond", "antique", "aquamarine", "azure", "beige", "bisque",
          "black", "blanched", "blue", "blush", "brown", "burlywood",
          "burnished", "chartreuse", "chiffon", "chocolate", "coral",
          "cornflower", "cornsilk", "cream", "cyan", "dark", "deep",
          "green", "goldenrod", "honeydew", "hot", "indian"]
WORDS = ["carefully", "quickly", "furiously", "slyly", "blithely", "even",
         "final", "ironic", "regular", "express", "bold", "pending",
         "deposits", "accounts", "packages", "theodolites", "instructions",
         "foxes", "pinto", "beans", "ideas", "requests"]

EPOCH_1992 = (pd.Timestamp("1992-01-01") - pd.Timestamp("1970-01-01")).days
DATE_RANGE = (pd.Timestamp("1998-08-02") - pd.Timestamp("1992-01-01")).days


def _date_col(idx, salt):
    days = EPOCH_1992 + _uni(idx, salt, 0, DATE_RANGE)
    return (days * 86400 * 10**9).view("datetime64[ns]")


def _comment_pool(n_pool, salt, trigger=None, trig_rate=0.012):
    rng = np.random.default_rng(salt)
    w = rng.choice(WORDS, size=(n_pool, 5))
    pool = np.array([" ".join(r) for r in w], dtype=object)
    if trigger:
        k = max(1, int(n_pool * trig_rate))
        hit = rng.choice(n_pool, k, replace=False)
        mids = rng.choice(WORDS, k)
        pool[hit] = [f"{trigger[0]} {m} {trigger[1]}" for m in mids]
    # de-duplicate (random sequences can collide; Categorical categories
    # must be unique) while keeping order
    uniq = list(dict.fromkeys(pool.tolist()))
    return np.array(uniq, dtype=object)


def _bounds(n, rank, world):
    base, rem = divmod(n, world)
    lo = rank * base + min(rank, rem)
    return lo, lo + base + (1 if rank < rem else 0)


ROWS = {
    "region": lambda sf: 5,
    "nation": lambda sf: 25,
    "supplier": lambda sf: max(1, int(10_000 * sf)),
    "customer": lambda sf: max(1, int(150_000 * sf)),
    "part": lambda sf: max(1, int(200_000 * sf)),
    "partsupp": lambda sf: max(4, int(800_000 * sf)),
    "orders": lambda sf: max(1, int(1_500_000 * sf)),
    "lineitem": lambda sf: 4 * max(1, int(1_500_000 * sf)),
}


def gen_table(name: str, sf: float, rank: int = 0, world: int = 1) -> pd.DataFrame:
    n = ROWS[name](sf)
    lo, hi = _bounds(n, rank, world)
    i = np.arange(lo, hi, dtype=np.uint64)
    key = i.astype(np.int64) + 1
    if name == "region":
        return pd.DataFrame({
            "R_REGIONKEY": np.arange(5, dtype=np.int64)[lo:hi],
            "R_NAME": np.array(REGIONS, dtype=object)[lo:hi],
            "R_COMMENT": np.array(["x"] * 5, dtype=object)[lo:hi],
        })
    if name == "nation":
        names = np.array([n0 for n0, _ in NATIONS], dtype=object)
        regs = np.array([r for _, r in NATIONS], dtype=np.int64)
        return pd.DataFrame({
            "N_NATIONKEY": np.arange(25, dtype=np.int64)[lo:hi],
            "N_NAME": names[lo:hi],
            "N_REGIONKEY": regs[lo:hi],
            "N_COMMENT": names[lo:hi],
        })
    if name == "supplier":
        pool = _comment_pool(5000, 42, ("Customer", "Complaints"))
        com = pd.Categorical.from_codes(
            _uni(i, 3, 0, len(pool) - 1).astype(np.int32), categories=list(pool))
        return pd.DataFrame({
            "S_SUPPKEY": key,
            "S_NAME": np.char.add("Supplier#",
                                  np.char.zfill(key.astype("U9"), 9)).astype(object),
            "S_ADDRESS": np.char.add("addr ", key.astype("U12")).astype(object),
            "S_NATIONKEY": _uni(i, 1, 0, 24),
            "S_PHONE": _phone(i, 1),
            "S_ACCTBAL": _unif(i, 2, -999.99, 9999.99).round(2),
            "S_COMMENT": com,
        })
    if name == "customer":
        nk = _uni(i, 10, 0, 24)
        return pd.DataFrame({
            "C_CUSTKEY": key,
            "C_NAME": np.char.add("Customer#",
                                  np.char.zfill(key.astype("U9"), 9)).astype(object),
            "C_ADDRESS": np.char.add("caddr ", key.astype("U12")).astype(object),
            "C_NATIONKEY": nk,
            "C_PHONE": _phone(i, 11, nk),
            "C_ACCTBAL": _unif(i, 12, -999.99, 9999.99).round(2),
            "C_MKTSEGMENT": _pick(i, 13, SEGMENTS),
            "C_COMMENT": np.char.add("com ", key.astype("U12")).astype(object),
        })
    if name == "part":
        # composite categoricals: build the small cross-product category
        # lists once and index them with combined codes
        ptype_cats = [f"{a} {b} {c}" for a in TYPE1 for b in TYPE2
                      for c in TYPE3]
        pt_code = (_uni(i, 20, 0, len(TYPE1) - 1) * len(TYPE2)
                   + _uni(i, 21, 0, len(TYPE2) - 1)) * len(TYPE3)             + _uni(i, 22, 0, len(TYPE3) - 1)
        ptype = pd.Categorical.from_codes(pt_code.astype(np.int32),
                                          categories=ptype_cats)
        cont_cats = [f"{a} {b}" for a in CONT1 for b in CONT2]
        ct_code = _uni(i, 23, 0, len(CONT1) - 1) * len(CONT2)             + _uni(i, 24, 0, len(CONT2) - 1)
        cont = pd.Categorical.from_codes(ct_code.astype(np.int32),
                                         categories=cont_cats)
        brand_cats = [f"Brand#{a}{b}" for a in range(1, 6)
                      for b in range(1, 6)]
        br_code = (_uni(i, 25, 1, 5) - 1) * 5 + (_uni(i, 26, 1, 5) - 1)
        brand = pd.Categorical.from_codes(br_code.astype(np.int32),
                                          categories=brand_cats)
        pname_cats = [f"{a} {b}" for a in COLORS for b in COLORS]
        pn_code = _uni(i, 27, 0, len(COLORS) - 1) * len(COLORS)             + _uni(i, 28, 0, len(COLORS) - 1)
        pname = pd.Categorical.from_codes(pn_code.astype(np.int32),
                                          categories=pname_cats)
        nm1 = pd.Categorical.from_codes(
            _uni(i, 27, 0, len(COLORS) - 1).astype(np.int32),
            categories=COLORS)
        return pd.DataFrame({
            "P_PARTKEY": key,
            "P_NAME": pname,
            "P_MFGR": pd.Categorical.from_codes(
                (_uni(i, 29, 1, 5) - 1).astype(np.int32),
                categories=[f"Manufacturer#{k}" for k in range(1, 6)]),
            "P_BRAND": brand,
            "P_TYPE": ptype,
            "P_SIZE": _uni(i, 30, 1, 50),
            "P_CONTAINER": cont,
            "P_RETAILPRICE": _unif(i, 31, 900.0, 2000.0).round(2),
            "P_COMMENT": nm1,
        })
    if name == "partsupp":
        nparts = ROWS["part"](sf)
        nsupp = ROWS["supplier"](sf)
        partkey = (i // 4).astype(np.int64) + 1
        j = (i % 4).astype(np.int64)
        suppkey = _ps_suppkey(partkey, j, nsupp)
        return pd.DataFrame({
            "PS_PARTKEY": partkey,
            "PS_SUPPKEY": suppkey,
            "PS_AVAILQTY": _uni(i, 40, 1, 9999),
            "PS_SUPPLYCOST": _unif(i, 41, 1.0, 1000.0).round(2),
            "PS_COMMENT": _pick(i, 42, WORDS),
        })
    if name == "orders":
        ncust = ROWS["customer"](sf)
        pool = _comment_pool(8000, 77, ("special", "requests"))
        com = pd.Categorical.from_codes(
            _uni(i, 56, 0, len(pool) - 1).astype(np.int32),
            categories=list(pool))
        return pd.DataFrame({
            "O_ORDERKEY": key,
            # custkeys divisible by 3 never order (q22's not-exists branch;
            # dbgen similarly skips a third of custkeys)
            "O_CUSTKEY": _no_mult3(_uni(i, 50, 1, ncust), ncust),
            "O_ORDERSTATUS": _pick(i, 51, ["F", "O", "P"]),
            "O_TOTALPRICE": _unif(i, 52, 1000.0, 500000.0).round(2),
            "O_ORDERDATE": _date_col(i, 53),
            "O_ORDERPRIORITY": _pick(i, 54, PRIORITIES),
            "O_CLERK": pd.Categorical.from_codes(
                (_uni(i, 55, 1, 1000) - 1).astype(np.int32),
                categories=[f"Clerk#{k}" for k in range(1, 1001)]),
            "O_SHIPPRIORITY": np.zeros(len(i), dtype=np.int64),
            "O_COMMENT": com,
        })
    if name == "lineitem":
        nparts = ROWS["part"](sf)
        nsupp = ROWS["supplier"](sf)
        norders = ROWS["orders"](sf)
        orderkey = (i // 4).astype(np.int64) + 1
        # a small share of lines concentrate on few orders so per-order
        # quantity sums exceed q18's 300 threshold (dbgen: 1-7 lines/order)
        hot = _mix(i, 90) % U64(50) == 0
        hot_key = (_mix(i, 91) % U64(max(1, norders // 200))).astype(np.int64) + 1
        orderkey = np.where(hot, hot_key, orderkey)
        partkey = _uni(i, 60, 1, nparts)
        suppkey = _ps_suppkey(partkey, _uni(i, 61, 0, 3), nsupp)
        qty = _uni(i, 62, 1, 50).astype(np.float64)
        price = _unif(i, 63, 900.0, 2000.0).round(2)
        ship = _date_col(i, 64)
        commit = ship + np.timedelta64(1, "D") * _uni(i, 65, -60, 60)
        receipt = ship + np.timedelta64(1, "D") * _uni(i, 66, 1, 30)
        return pd.DataFrame({
            "L_ORDERKEY": orderkey,
            "L_PARTKEY": partkey,
            "L_SUPPKEY": suppkey,
            "L_LINENUMBER": (i % 4).astype(np.int64) + 1,
            "L_QUANTITY": qty,
            "L_EXTENDEDPRICE": (qty * price).round(2),
            "L_DISCOUNT": (_uni(i, 67, 0, 10) / 100.0),
            "L_TAX": (_uni(i, 68, 0, 8) / 100.0),
            "L_RETURNFLAG": _pick(i, 69, ["A", "N", "R"]),
            "L_LINESTATUS": _pick(i, 70, ["O", "F"]),
            "L_SHIPDATE": ship,
            "L_COMMITDATE": commit,
            "L_RECEIPTDATE": receipt,
            "L_SHIPINSTRUCT": _pick(i, 71, SHIPINSTRUCT),
            "L_SHIPMODE": _pick(i, 72, SHIPMODES),
            "L_COMMENT": _pick(i, 73, WORDS),
        })
    raise KeyError(name)


def _no_mult3(ck, ncust):
    ck = np.where(ck % 3 == 0, ck + 1, ck)
    return np.where(ck > ncust, 1, ck)


def _phone(i, salt, nationkey=None):
    cc = (10 + (nationkey if nationkey is not None else _uni(i, salt, 0, 24)))
    p1 = _uni(i, salt + 100, 100, 999)
    p2 = _uni(i, salt + 101, 100, 999)
    p3 = _uni(i, salt + 102, 1000, 9999)
    out = np.char.add(np.char.add(np.char.add(np.char.add(
        np.char.add(np.char.add(cc.astype("U2"), "-"), p1.astype("U3")), "-"),
        p2.astype("U3")), "-"), p3.astype("U4"))
    return out.astype(object)


def _ps_suppkey(partkey, j, nsupp):
    return ((partkey + j * (nsupp // 4 + 1)) % nsupp) + 1


TABLES = ["region", "nation", "supplier", "customer", "part", "partsupp",
          "orders", "lineitem"]


def gen_arrow(name: str, sf: float, rank: int = 0, world: int = 1):
    """Arrow table without pandas block consolidation (2x faster, no 2D
    stacking copies) — the path the GPU benchmarks load through."""
    import pyarrow as pa

    df = gen_table(name, sf, rank, world)
    return pa.Table.from_pandas(df, preserve_index=False)


def gen_all(sf: float, rank: int = 0, world: int = 1):
    return {t: gen_table(t, sf, rank, world) for t in TABLES}
