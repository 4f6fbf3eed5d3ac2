"""TPC-H Q1-Q22 in pandas form, written for the bodo_amd.pandas lazy API
(also runs on plain pandas for differential testing).

These are original implementations of the standard TPC-H pandas
formulations (the public polars-benchmark / coiled-benchmarks shapes the
reference also derives from, benchmarks/tpch/README.md).  Each query takes
(m, t): m = pandas-like module, t = dict of table frames.
"""

import pandas as pd

TS = pd.Timestamp


def q1(m, t):
    li = t["lineitem"]
    f = li[li.L_SHIPDATE <= TS("1998-09-02")]
    f["DISC_PRICE"] = f.L_EXTENDEDPRICE * (1.0 - f.L_DISCOUNT)
    f["CHARGE"] = f.L_EXTENDEDPRICE * (1.0 - f.L_DISCOUNT) * (1.0 + f.L_TAX)
    g = f.groupby(["L_RETURNFLAG", "L_LINESTATUS"], as_index=False).agg(
        SUM_QTY=m.NamedAgg("L_QUANTITY", "sum"),
        SUM_BASE_PRICE=m.NamedAgg("L_EXTENDEDPRICE", "sum"),
        SUM_DISC_PRICE=m.NamedAgg("DISC_PRICE", "sum"),
        SUM_CHARGE=m.NamedAgg("CHARGE", "sum"),
        AVG_QTY=m.NamedAgg("L_QUANTITY", "mean"),
        AVG_PRICE=m.NamedAgg("L_EXTENDEDPRICE", "mean"),
        AVG_DISC=m.NamedAgg("L_DISCOUNT", "mean"),
        COUNT_ORDER=m.NamedAgg("L_ORDERKEY", "size"))
    return g.sort_values(["L_RETURNFLAG", "L_LINESTATUS"])


def q2(m, t):
    jn = (t["part"]
          .merge(t["partsupp"], left_on="P_PARTKEY", right_on="PS_PARTKEY")
          .merge(t["supplier"], left_on="PS_SUPPKEY", right_on="S_SUPPKEY")
          .merge(t["nation"], left_on="S_NATIONKEY", right_on="N_NATIONKEY")
          .merge(t["region"], left_on="N_REGIONKEY", right_on="R_REGIONKEY"))
    jn = jn[(jn.P_SIZE == 15)]
    jn = jn[jn.P_TYPE.str.endswith("BRASS")]
    jn = jn[jn.R_NAME == "EUROPE"]
    best = jn.groupby("P_PARTKEY", as_index=False)["PS_SUPPLYCOST"].min()
    jn2 = best.merge(jn, on=["P_PARTKEY", "PS_SUPPLYCOST"])
    out = jn2.loc[:, ["S_ACCTBAL", "S_NAME", "N_NAME", "P_PARTKEY",
                      "P_MFGR", "S_ADDRESS", "S_PHONE", "S_COMMENT"]]
    return out.sort_values(["S_ACCTBAL", "N_NAME", "S_NAME", "P_PARTKEY"],
                           ascending=[False, True, True, True]).head(100)


def q3(m, t):
    var = TS("1995-03-15")
    cust = t["customer"]
    cust = cust[cust.C_MKTSEGMENT == "BUILDING"]
    jn = cust.merge(t["orders"], left_on="C_CUSTKEY", right_on="O_CUSTKEY") \
             .merge(t["lineitem"], left_on="O_ORDERKEY", right_on="L_ORDERKEY")
    jn = jn[jn.O_ORDERDATE < var]
    jn = jn[jn.L_SHIPDATE > var]
    jn["REVENUE"] = jn.L_EXTENDEDPRICE * (1 - jn.L_DISCOUNT)
    g = jn.groupby(["O_ORDERKEY", "O_ORDERDATE", "O_SHIPPRIORITY"],
                   as_index=False)["REVENUE"].sum()
    g = g.loc[:, ["O_ORDERKEY", "REVENUE", "O_ORDERDATE", "O_SHIPPRIORITY"]]
    g = g.rename(columns={"O_ORDERKEY": "L_ORDERKEY"})
    return g.sort_values(["REVENUE", "O_ORDERDATE"],
                         ascending=[False, True]).head(10)


def q4(m, t):
    li, orders = t["lineitem"], t["orders"]
    fl = li[li.L_COMMITDATE < li.L_RECEIPTDATE]
    fo = orders[(orders.O_ORDERDATE >= TS("1993-08-01"))
                & (orders.O_ORDERDATE < TS("1993-11-01"))]
    jn = fo[fo.O_ORDERKEY.isin(fl.L_ORDERKEY)]
    out = jn.groupby("O_ORDERPRIORITY", as_index=False)["O_ORDERKEY"].count() \
            .sort_values(["O_ORDERPRIORITY"])
    out.columns = ["O_ORDERPRIORITY", "ORDER_COUNT"]
    return out


def q5(m, t):
    jn = (t["region"]
          .merge(t["nation"], left_on="R_REGIONKEY", right_on="N_REGIONKEY")
          .merge(t["customer"], left_on="N_NATIONKEY", right_on="C_NATIONKEY")
          .merge(t["orders"], left_on="C_CUSTKEY", right_on="O_CUSTKEY")
          .merge(t["lineitem"], left_on="O_ORDERKEY", right_on="L_ORDERKEY")
          .merge(t["supplier"], left_on=["L_SUPPKEY", "N_NATIONKEY"],
                 right_on=["S_SUPPKEY", "S_NATIONKEY"]))
    jn = jn[jn.R_NAME == "ASIA"]
    jn = jn[(jn.O_ORDERDATE >= TS("1996-01-01"))
            & (jn.O_ORDERDATE < TS("1997-01-01"))]
    jn["REVENUE"] = jn.L_EXTENDEDPRICE * (1.0 - jn.L_DISCOUNT)
    g = jn.groupby("N_NAME", as_index=False)["REVENUE"].sum()
    return g.sort_values("REVENUE", ascending=False)


def q6(m, t):
    li = t["lineitem"]
    f = li[(li.L_SHIPDATE >= TS("1996-01-01"))
           & (li.L_SHIPDATE < TS("1997-01-01"))]
    f = f[(f.L_DISCOUNT >= 0.08) & (f.L_DISCOUNT <= 0.1)]
    f = f[f.L_QUANTITY < 24]
    rev = (f.L_EXTENDEDPRICE * f.L_DISCOUNT).sum()
    return m.DataFrame({"REVENUE": [rev]})


def q7(m, t):
    n1 = t["nation"].rename(columns={"N_NATIONKEY": "N1_KEY",
                                     "N_NAME": "SUPP_NATION"})
    n2 = t["nation"].rename(columns={"N_NATIONKEY": "N2_KEY",
                                     "N_NAME": "CUST_NATION"})
    jn = (t["supplier"]
          .merge(t["lineitem"], left_on="S_SUPPKEY", right_on="L_SUPPKEY")
          .merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY")
          .merge(t["customer"], left_on="O_CUSTKEY", right_on="C_CUSTKEY")
          .merge(n1[["N1_KEY", "SUPP_NATION"]], left_on="S_NATIONKEY",
                 right_on="N1_KEY")
          .merge(n2[["N2_KEY", "CUST_NATION"]], left_on="C_NATIONKEY",
                 right_on="N2_KEY"))
    jn = jn[((jn.SUPP_NATION == "FRANCE") & (jn.CUST_NATION == "GERMANY"))
            | ((jn.SUPP_NATION == "GERMANY") & (jn.CUST_NATION == "FRANCE"))]
    jn = jn[(jn.L_SHIPDATE >= TS("1995-01-01"))
            & (jn.L_SHIPDATE <= TS("1996-12-31"))]
    jn["L_YEAR"] = jn.L_SHIPDATE.dt.year
    jn["VOLUME"] = jn.L_EXTENDEDPRICE * (1.0 - jn.L_DISCOUNT)
    g = jn.groupby(["SUPP_NATION", "CUST_NATION", "L_YEAR"],
                   as_index=False)["VOLUME"].sum()
    g = g.rename(columns={"VOLUME": "REVENUE"})
    return g.sort_values(["SUPP_NATION", "CUST_NATION", "L_YEAR"])


def q8(m, t):
    n1 = t["nation"].loc[:, ["N_NATIONKEY", "N_REGIONKEY"]]
    n2 = t["nation"].loc[:, ["N_NATIONKEY", "N_NAME"]]
    jn = (t["part"]
          .merge(t["lineitem"], left_on="P_PARTKEY", right_on="L_PARTKEY")
          .merge(t["supplier"], left_on="L_SUPPKEY", right_on="S_SUPPKEY")
          .merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY")
          .merge(t["customer"], left_on="O_CUSTKEY", right_on="C_CUSTKEY")
          .merge(n1, left_on="C_NATIONKEY", right_on="N_NATIONKEY")
          .merge(t["region"], left_on="N_REGIONKEY", right_on="R_REGIONKEY"))
    jn = jn[jn.R_NAME == "AMERICA"]
    jn = jn.merge(n2, left_on="S_NATIONKEY", right_on="N_NATIONKEY")
    jn = jn[(jn.O_ORDERDATE >= TS("1995-01-01"))
            & (jn.O_ORDERDATE < TS("1997-01-01"))]
    jn = jn[jn.P_TYPE == "ECONOMY ANODIZED STEEL"]
    jn["O_YEAR"] = jn.O_ORDERDATE.dt.year
    jn["VOLUME"] = jn.L_EXTENDEDPRICE * (1.0 - jn.L_DISCOUNT)
    jn = jn.rename(columns={"N_NAME": "NATION"})
    denom = jn.groupby("O_YEAR", as_index=False)["VOLUME"].sum() \
              .rename(columns={"VOLUME": "TOTAL"})
    num = jn[jn.NATION == "BRAZIL"] \
        .groupby("O_YEAR", as_index=False)["VOLUME"].sum() \
        .rename(columns={"VOLUME": "BRAZIL_VOL"})
    agg = denom.merge(num, on="O_YEAR", how="left")
    agg["MKT_SHARE"] = (agg.BRAZIL_VOL / agg.TOTAL).round(2)
    return agg.sort_values("O_YEAR")[["O_YEAR", "MKT_SHARE"]]


def q9(m, t):
    jn = (t["lineitem"]
          .merge(t["supplier"], left_on="L_SUPPKEY", right_on="S_SUPPKEY")
          .merge(t["partsupp"], left_on=["L_SUPPKEY", "L_PARTKEY"],
                 right_on=["PS_SUPPKEY", "PS_PARTKEY"])
          .merge(t["part"], left_on="L_PARTKEY", right_on="P_PARTKEY")
          .merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY")
          .merge(t["nation"], left_on="S_NATIONKEY", right_on="N_NATIONKEY"))
    jn = jn[jn.P_NAME.str.contains("green")]
    jn["O_YEAR"] = jn.O_ORDERDATE.dt.year
    jn["AMOUNT"] = (jn.L_EXTENDEDPRICE * (1 - jn.L_DISCOUNT)
                    - jn.PS_SUPPLYCOST * jn.L_QUANTITY)
    g = jn.groupby(["N_NAME", "O_YEAR"], as_index=False)["AMOUNT"].sum()
    g = g.rename(columns={"N_NAME": "NATION", "AMOUNT": "SUM_PROFIT"})
    return g.sort_values(["NATION", "O_YEAR"], ascending=[True, False])


def q10(m, t):
    orders = t["orders"]
    fo = orders[(orders.O_ORDERDATE >= TS("1994-11-01"))
                & (orders.O_ORDERDATE < TS("1995-02-01"))]
    li = t["lineitem"]
    fl = li[li.L_RETURNFLAG == "R"]
    jn = fl.merge(fo, left_on="L_ORDERKEY", right_on="O_ORDERKEY") \
           .merge(t["customer"], left_on="O_CUSTKEY", right_on="C_CUSTKEY") \
           .merge(t["nation"], left_on="C_NATIONKEY", right_on="N_NATIONKEY")
    jn["REVENUE"] = jn.L_EXTENDEDPRICE * (1.0 - jn.L_DISCOUNT)
    g = jn.groupby(["C_CUSTKEY", "C_NAME", "C_ACCTBAL", "C_PHONE", "N_NAME",
                    "C_ADDRESS", "C_COMMENT"], as_index=False)["REVENUE"].sum()
    g["REVENUE"] = g.REVENUE.round(2)
    return g.sort_values("REVENUE", ascending=False).head(20)


def q11(m, t):
    jn = (t["partsupp"]
          .merge(t["supplier"], left_on="PS_SUPPKEY", right_on="S_SUPPKEY")
          .merge(t["nation"], left_on="S_NATIONKEY", right_on="N_NATIONKEY"))
    jn = jn[jn.N_NAME == "GERMANY"]
    jn["VALUE"] = jn.PS_SUPPLYCOST * jn.PS_AVAILQTY
    threshold = jn["VALUE"].sum() * 0.0001
    g = jn.groupby("PS_PARTKEY", as_index=False)["VALUE"].sum()
    g = g[g.VALUE > threshold]
    return g.sort_values("VALUE", ascending=False)


def q12(m, t):
    li = t["lineitem"]
    f = li[li.L_SHIPMODE.isin(("MAIL", "SHIP"))]
    f = f[f.L_COMMITDATE < f.L_RECEIPTDATE]
    f = f[f.L_SHIPDATE < f.L_COMMITDATE]
    f = f[(f.L_RECEIPTDATE >= TS("1994-01-01"))
          & (f.L_RECEIPTDATE < TS("1995-01-01"))]
    jn = f.merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY")
    jn["HIGH"] = jn.O_ORDERPRIORITY.isin(("1-URGENT", "2-HIGH")) \
        .astype("int64")
    jn["LOW"] = (~jn.O_ORDERPRIORITY.isin(("1-URGENT", "2-HIGH"))) \
        .astype("int64")
    g = jn.groupby("L_SHIPMODE", as_index=False).agg(
        HIGH_LINE_COUNT=m.NamedAgg("HIGH", "sum"),
        LOW_LINE_COUNT=m.NamedAgg("LOW", "sum"))
    return g.sort_values("L_SHIPMODE")


def q13(m, t):
    orders = t["orders"]
    fo = orders[~orders.O_COMMENT.str.contains("special.*requests")]
    jn = t["customer"].merge(fo, left_on="C_CUSTKEY", right_on="O_CUSTKEY",
                             how="left")
    g1 = jn.groupby("C_CUSTKEY", as_index=False).agg(
        C_COUNT=m.NamedAgg("O_ORDERKEY", "count"))
    g2 = g1.groupby("C_COUNT", as_index=False).agg(
        CUSTDIST=m.NamedAgg("C_CUSTKEY", "size"))
    return g2.sort_values(["CUSTDIST", "C_COUNT"], ascending=[False, False])


def q14(m, t):
    li = t["lineitem"]
    f = li[(li.L_SHIPDATE >= TS("1994-03-01"))
           & (li.L_SHIPDATE < TS("1994-04-01"))]
    jn = f.merge(t["part"], left_on="L_PARTKEY", right_on="P_PARTKEY")
    jn["REV"] = jn.L_EXTENDEDPRICE * (1 - jn.L_DISCOUNT)
    jn["PROMO_REV"] = jn.REV.where(jn.P_TYPE.str.startswith("PROMO"), 0.0)
    ratio = 100.0 * jn["PROMO_REV"].sum() / jn["REV"].sum()
    return m.DataFrame({"PROMO_REVENUE": [round(ratio, 2)]})


def q15(m, t):
    li = t["lineitem"]
    f = li[(li.L_SHIPDATE >= TS("1996-01-01"))
           & (li.L_SHIPDATE < TS("1996-04-01"))]
    f["REVENUE"] = f.L_EXTENDEDPRICE * (1 - f.L_DISCOUNT)
    rev = f.groupby("L_SUPPKEY", as_index=False).agg(
        TOTAL_REVENUE=m.NamedAgg("REVENUE", "sum"))
    rev = rev.rename(columns={"L_SUPPKEY": "SUPPLIER_NO"})
    jn = t["supplier"].merge(rev, left_on="S_SUPPKEY",
                             right_on="SUPPLIER_NO", how="inner")
    mx = rev["TOTAL_REVENUE"].max()
    jn = jn[jn.TOTAL_REVENUE == mx]
    return jn[["S_SUPPKEY", "S_NAME", "S_ADDRESS", "S_PHONE",
               "TOTAL_REVENUE"]].sort_values("S_SUPPKEY")


def q16(m, t):
    supp = t["supplier"]
    bad = supp[supp.S_COMMENT.str.contains("Customer.*Complaints")]["S_SUPPKEY"]
    ps = t["partsupp"]
    ps = ps[~ps.PS_SUPPKEY.isin(bad)]
    jn = ps.merge(t["part"], left_on="PS_PARTKEY", right_on="P_PARTKEY")
    jn = jn[(jn.P_BRAND != "Brand#45")
            & (~jn.P_TYPE.str.startswith("MEDIUM POLISHED"))
            & (jn.P_SIZE.isin((49, 14, 23, 45, 19, 3, 36, 9)))]
    g = jn.groupby(["P_BRAND", "P_TYPE", "P_SIZE"],
                   as_index=False)["PS_SUPPKEY"].nunique()
    g = g.rename(columns={"PS_SUPPKEY": "SUPPLIER_CNT"})
    return g.sort_values(["SUPPLIER_CNT", "P_BRAND", "P_TYPE", "P_SIZE"],
                         ascending=[False, True, True, True])


def q17(m, t):
    jn = t["lineitem"].merge(t["part"], left_on="L_PARTKEY",
                             right_on="P_PARTKEY")
    jn = jn[(jn.P_BRAND == "Brand#23") & (jn.P_CONTAINER == "MED BOX")]
    avg = jn.groupby("L_PARTKEY", as_index=False).agg(
        QTY_AVG=m.NamedAgg("L_QUANTITY", "mean"))
    jn2 = jn.merge(avg, on="L_PARTKEY", how="left")
    jn2 = jn2[jn2.L_QUANTITY < 0.2 * jn2.QTY_AVG]
    total = jn2["L_EXTENDEDPRICE"].sum() / 7.0
    return m.DataFrame({"AVG_YEARLY": [round(total, 2)]})


def q18(m, t):
    g1 = t["lineitem"].groupby("L_ORDERKEY", as_index=False,
                               sort=False)["L_QUANTITY"].sum()
    f = g1[g1.L_QUANTITY > 300]
    jn = f.merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY") \
          .merge(t["customer"], left_on="O_CUSTKEY", right_on="C_CUSTKEY")
    g2 = jn.groupby(["C_NAME", "C_CUSTKEY", "O_ORDERKEY", "O_ORDERDATE",
                     "O_TOTALPRICE"], as_index=False, sort=False)[
        "L_QUANTITY"].sum()
    return g2.sort_values(["O_TOTALPRICE", "O_ORDERDATE"],
                          ascending=[False, True]).head(100)


def q19(m, t):
    jn = t["lineitem"].merge(t["part"], left_on="L_PARTKEY",
                             right_on="P_PARTKEY")
    sm = ((jn.P_BRAND == "Brand#31")
          & (jn.P_CONTAINER.isin(("SM CASE", "SM BOX", "SM PACK", "SM PKG")))
          & ((jn.L_QUANTITY >= 4) & (jn.L_QUANTITY <= 14))
          & (jn.P_SIZE <= 5))
    med = ((jn.P_BRAND == "Brand#43")
           & (jn.P_CONTAINER.isin(("MED BAG", "MED BOX", "MED PKG",
                                   "MED PACK")))
           & ((jn.L_QUANTITY >= 15) & (jn.L_QUANTITY <= 25))
           & ((jn.P_SIZE >= 1) & (jn.P_SIZE <= 10)))
    lg = ((jn.P_BRAND == "Brand#43")
          & (jn.P_CONTAINER.isin(("LG CASE", "LG BOX", "LG PACK", "LG PKG")))
          & ((jn.L_QUANTITY >= 26) & (jn.L_QUANTITY <= 36))
          & (jn.P_SIZE <= 15))
    common = (jn.L_SHIPMODE.isin(("AIR", "AIR REG"))) \
        & (jn.L_SHIPINSTRUCT == "DELIVER IN PERSON")
    f = jn[(sm | med | lg) & common]
    rev = (f.L_EXTENDEDPRICE * (1 - f.L_DISCOUNT)).sum()
    return m.DataFrame({"REVENUE": [rev]})


def q20(m, t):
    li = t["lineitem"]
    f = li[(li.L_SHIPDATE >= TS("1996-01-01"))
           & (li.L_SHIPDATE < TS("1997-01-01"))]
    agg = f.groupby(["L_SUPPKEY", "L_PARTKEY"], as_index=False).agg(
        SUM_QTY=m.NamedAgg("L_QUANTITY", "sum"))
    agg["SUM_QTY"] = agg.SUM_QTY * 0.5
    fn = t["nation"]
    fn = fn[fn.N_NAME == "JORDAN"]
    jn1 = t["supplier"].merge(fn, left_on="S_NATIONKEY",
                              right_on="N_NATIONKEY")
    part = t["part"]
    fp = part[part.P_NAME.str.startswith("azure")]
    jn2 = t["partsupp"].merge(fp, left_on="PS_PARTKEY", right_on="P_PARTKEY")
    jn3 = jn2.merge(agg, left_on=["PS_SUPPKEY", "PS_PARTKEY"],
                    right_on=["L_SUPPKEY", "L_PARTKEY"])
    jn3 = jn3[jn3.PS_AVAILQTY > jn3.SUM_QTY]
    jn4 = jn1.merge(jn3, left_on="S_SUPPKEY", right_on="PS_SUPPKEY")
    return jn4[["S_NAME", "S_ADDRESS"]].sort_values("S_NAME")


def q21(m, t):
    li = t["lineitem"]
    g1 = li.groupby("L_ORDERKEY", as_index=False).agg(
        NSUPP=m.NamedAgg("L_SUPPKEY", "nunique"))
    g1 = g1[g1.NSUPP > 1]
    fl = li[li.L_RECEIPTDATE > li.L_COMMITDATE]
    jn1 = g1.merge(fl, on="L_ORDERKEY")
    g2 = jn1.groupby("L_ORDERKEY", as_index=False).agg(
        NSUPP2=m.NamedAgg("L_SUPPKEY", "nunique"))
    jn = g2.merge(jn1, on="L_ORDERKEY") \
        .merge(t["orders"], left_on="L_ORDERKEY", right_on="O_ORDERKEY") \
        .merge(t["supplier"], left_on="L_SUPPKEY", right_on="S_SUPPKEY") \
        .merge(t["nation"], left_on="S_NATIONKEY", right_on="N_NATIONKEY")
    jn = jn[(jn.NSUPP2 == 1) & (jn.N_NAME == "SAUDI ARABIA")
            & (jn.O_ORDERSTATUS == "F")]
    g3 = jn.groupby("S_NAME", as_index=False).agg(
        NUMWAIT=m.NamedAgg("NSUPP2", "size"))
    return g3.sort_values(["NUMWAIT", "S_NAME"],
                          ascending=[False, True]).head(100)


def q22(m, t):
    cust = t["customer"]
    cust["CNTRYCODE"] = cust.C_PHONE.str.strip().str.slice(0, 2)
    f = cust[cust.CNTRYCODE.isin(("13", "31", "23", "29", "30", "18", "17"))]
    pos = f[f.C_ACCTBAL > 0.0]
    avg_bal = pos["C_ACCTBAL"].mean()
    rich = f[f.C_ACCTBAL > avg_bal]
    jn = rich.merge(t["orders"], left_on="C_CUSTKEY", right_on="O_CUSTKEY",
                    how="left")
    jn = jn[jn.O_CUSTKEY.isnull()]
    g = jn.groupby("CNTRYCODE", as_index=False).agg(
        NUMCUST=m.NamedAgg("C_ACCTBAL", "size"),
        TOTACCTBAL=m.NamedAgg("C_ACCTBAL", "sum"))
    return g.sort_values("CNTRYCODE")


ALL = {i: globals()[f"q{i}"] for i in range(1, 23)}
