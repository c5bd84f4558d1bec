"""TPC-H correctness: all 22 queries vs independent pandas oracles on the
same generated tables (the reference validates against precomputed answers;
we validate against a second implementation — tests/conftest pattern)."""
import datetime as dt

import numpy as np
import pandas as pd
import pytest

from benchmarks.tpch import datagen, queries

SF = 0.02


@pytest.fixture(scope="module")
def tables():
    return datagen.dataframes(SF, device="cpu")


@pytest.fixture(scope="module")
def pdf(tables):
    return {k: v.to_pandas() for k, v in tables.items()}


def run(i, tables):
    return queries.run_query(i, tables, sf=SF).to_pandas()


def assert_frames(actual: pd.DataFrame, expected: pd.DataFrame,
                  sort_cols=None, float_cols=(), check_row_order=True):
    assert list(actual.columns) == list(expected.columns), \
        (list(actual.columns), list(expected.columns))
    a, e = actual.copy(), expected.copy()
    if sort_cols:
        a = a.sort_values(sort_cols).reset_index(drop=True)
        e = e.sort_values(sort_cols).reset_index(drop=True)
    else:
        a = a.reset_index(drop=True)
        e = e.reset_index(drop=True)
    assert len(a) == len(e), f"{len(a)} rows vs {len(e)}"
    for c in a.columns:
        av, ev = a[c].to_numpy(), e[c].to_numpy()
        if c in float_cols or a[c].dtype.kind == "f":
            np.testing.assert_allclose(av.astype(float), ev.astype(float),
                                       rtol=1e-9, atol=1e-6, err_msg=c)
        else:
            assert list(av) == list(ev), f"column {c}"


def _rev(df):
    return df["l_extendedprice"] * (1 - df["l_discount"])


def test_q1(tables, pdf):
    li = pdf["lineitem"]
    f = li[li.l_shipdate <= dt.date(1998, 9, 2)].copy()
    f["disc_price"] = _rev(f)
    f["charge"] = f.disc_price * (1 + f.l_tax)
    g = f.groupby(["l_returnflag", "l_linestatus"], as_index=False).agg(
        sum_qty=("l_quantity", "sum"),
        sum_base_price=("l_extendedprice", "sum"),
        sum_disc_price=("disc_price", "sum"),
        sum_charge=("charge", "sum"),
        avg_qty=("l_quantity", "mean"),
        avg_price=("l_extendedprice", "mean"),
        avg_disc=("l_discount", "mean"),
        count_order=("l_quantity", "count"),
    ).sort_values(["l_returnflag", "l_linestatus"]).reset_index(drop=True)
    got = run(1, tables)
    g["count_order"] = g["count_order"].astype("uint64")
    assert_frames(got, g)


def test_q2(tables, pdf):
    part = pdf["part"]
    part = part[(part.p_size == 15) & part.p_type.str.endswith("BRASS")]
    nat = pdf["nation"].merge(pdf["region"][pdf["region"].r_name == "EUROPE"],
                              left_on="n_regionkey", right_on="r_regionkey")
    supp = pdf["supplier"].merge(nat, left_on="s_nationkey",
                                 right_on="n_nationkey")
    ps = pdf["partsupp"].merge(part, left_on="ps_partkey",
                               right_on="p_partkey") \
        .merge(supp, left_on="ps_suppkey", right_on="s_suppkey")
    mins = ps.groupby("ps_partkey", as_index=False).agg(
        min_cost=("ps_supplycost", "min"))
    out = ps.merge(mins, on="ps_partkey")
    out = out[out.ps_supplycost == out.min_cost]
    out = out.drop(columns=["p_partkey"]) \
        .rename(columns={"ps_partkey": "p_partkey"})[
        ["s_acctbal", "s_name", "n_name", "p_partkey", "p_mfgr", "s_address",
         "s_phone", "s_comment"]]
    out = out.sort_values(["s_acctbal", "n_name", "s_name", "p_partkey"],
                          ascending=[False, True, True, True]).head(100) \
        .reset_index(drop=True)
    got = run(2, tables)
    assert_frames(got, out)


def test_q3(tables, pdf):
    cust = pdf["customer"][pdf["customer"].c_mktsegment == "BUILDING"]
    orders = pdf["orders"][pdf["orders"].o_orderdate < dt.date(1995, 3, 15)]
    li = pdf["lineitem"][pdf["lineitem"].l_shipdate > dt.date(1995, 3, 15)]
    j = li.merge(orders, left_on="l_orderkey", right_on="o_orderkey") \
        .merge(cust, left_on="o_custkey", right_on="c_custkey")
    j["revenue"] = _rev(j)
    g = j.groupby(["l_orderkey", "o_orderdate", "o_shippriority"],
                  as_index=False).agg(revenue=("revenue", "sum"))
    g = g.rename(columns={"l_orderkey": "o_orderkey"})[
        ["o_orderkey", "revenue", "o_orderdate", "o_shippriority"]]
    g = g.sort_values(["revenue", "o_orderdate"],
                      ascending=[False, True]).head(10).reset_index(drop=True)
    got = run(3, tables)
    assert_frames(got, g, sort_cols=["revenue", "o_orderkey"])


def test_q4(tables, pdf):
    orders = pdf["orders"]
    orders = orders[(orders.o_orderdate >= dt.date(1993, 7, 1)) &
                    (orders.o_orderdate < dt.date(1993, 10, 1))]
    li = pdf["lineitem"]
    late = li[li.l_commitdate < li.l_receiptdate]
    sel = orders[orders.o_orderkey.isin(late.l_orderkey)]
    g = sel.groupby("o_orderpriority", as_index=False).agg(
        order_count=("o_orderkey", "count")).sort_values("o_orderpriority") \
        .reset_index(drop=True)
    g["order_count"] = g["order_count"].astype("uint64")
    got = run(4, tables)
    assert_frames(got, g)


def test_q5(tables, pdf):
    nat = pdf["nation"].merge(pdf["region"][pdf["region"].r_name == "ASIA"],
                              left_on="n_regionkey", right_on="r_regionkey")
    orders = pdf["orders"]
    orders = orders[(orders.o_orderdate >= dt.date(1994, 1, 1)) &
                    (orders.o_orderdate < dt.date(1995, 1, 1))]
    j = orders.merge(pdf["customer"], left_on="o_custkey",
                     right_on="c_custkey") \
        .merge(pdf["lineitem"], left_on="o_orderkey", right_on="l_orderkey") \
        .merge(pdf["supplier"], left_on=["l_suppkey", "c_nationkey"],
               right_on=["s_suppkey", "s_nationkey"]) \
        .merge(nat, left_on="c_nationkey", right_on="n_nationkey")
    j["revenue"] = _rev(j)
    g = j.groupby("n_name", as_index=False).agg(revenue=("revenue", "sum")) \
        .sort_values("revenue", ascending=False).reset_index(drop=True)
    got = run(5, tables)
    assert_frames(got, g)


def test_q6(tables, pdf):
    li = pdf["lineitem"]
    f = li[(li.l_shipdate >= dt.date(1994, 1, 1)) &
           (li.l_shipdate < dt.date(1995, 1, 1)) &
           (li.l_discount >= 0.05) & (li.l_discount <= 0.07) &
           (li.l_quantity < 24)]
    want = (f.l_extendedprice * f.l_discount).sum()
    got = run(6, tables)
    np.testing.assert_allclose(got["revenue"][0], want, rtol=1e-9)


def test_q7(tables, pdf):
    li = pdf["lineitem"]
    li = li[(li.l_shipdate >= dt.date(1995, 1, 1)) &
            (li.l_shipdate <= dt.date(1996, 12, 31))]
    j = li.merge(pdf["supplier"], left_on="l_suppkey", right_on="s_suppkey") \
        .merge(pdf["orders"], left_on="l_orderkey", right_on="o_orderkey") \
        .merge(pdf["customer"], left_on="o_custkey", right_on="c_custkey") \
        .merge(pdf["nation"].rename(columns={"n_nationkey": "n1_key",
                                             "n_name": "supp_nation"})
               [["n1_key", "supp_nation"]],
               left_on="s_nationkey", right_on="n1_key") \
        .merge(pdf["nation"].rename(columns={"n_nationkey": "n2_key",
                                             "n_name": "cust_nation"})
               [["n2_key", "cust_nation"]],
               left_on="c_nationkey", right_on="n2_key")
    j = j[((j.supp_nation == "FRANCE") & (j.cust_nation == "GERMANY")) |
          ((j.supp_nation == "GERMANY") & (j.cust_nation == "FRANCE"))]
    j["l_year"] = pd.to_datetime(j.l_shipdate).dt.year.astype("int32")
    j["revenue"] = _rev(j)
    g = j.groupby(["supp_nation", "cust_nation", "l_year"],
                  as_index=False).agg(revenue=("revenue", "sum")) \
        .sort_values(["supp_nation", "cust_nation", "l_year"]) \
        .reset_index(drop=True)
    got = run(7, tables)
    assert_frames(got, g)


def test_q8(tables, pdf):
    nat1 = pdf["nation"].merge(
        pdf["region"][pdf["region"].r_name == "AMERICA"],
        left_on="n_regionkey", right_on="r_regionkey")
    part = pdf["part"][pdf["part"].p_type == "ECONOMY ANODIZED STEEL"]
    orders = pdf["orders"]
    orders = orders[(orders.o_orderdate >= dt.date(1995, 1, 1)) &
                    (orders.o_orderdate <= dt.date(1996, 12, 31))]
    j = pdf["lineitem"] \
        .merge(part, left_on="l_partkey", right_on="p_partkey") \
        .merge(orders, left_on="l_orderkey", right_on="o_orderkey") \
        .merge(pdf["customer"], left_on="o_custkey", right_on="c_custkey") \
        .merge(nat1[["n_nationkey"]], left_on="c_nationkey",
               right_on="n_nationkey") \
        .merge(pdf["supplier"], left_on="l_suppkey", right_on="s_suppkey") \
        .merge(pdf["nation"].rename(columns={"n_nationkey": "n2_key",
                                             "n_name": "supp_nation"})
               [["n2_key", "supp_nation"]],
               left_on="s_nationkey", right_on="n2_key")
    j["o_year"] = pd.to_datetime(j.o_orderdate).dt.year.astype("int32")
    j["volume"] = _rev(j)
    j["bz"] = np.where(j.supp_nation == "BRAZIL", j.volume, 0.0)
    g = j.groupby("o_year", as_index=False).agg(num=("bz", "sum"),
                                                den=("volume", "sum"))
    g["mkt_share"] = g.num / g.den
    g = g[["o_year", "mkt_share"]].sort_values("o_year") \
        .reset_index(drop=True)
    got = run(8, tables)
    assert_frames(got, g)


def test_q9(tables, pdf):
    part = pdf["part"][pdf["part"].p_name.str.contains("green")]
    j = pdf["lineitem"] \
        .merge(part, left_on="l_partkey", right_on="p_partkey") \
        .merge(pdf["supplier"], left_on="l_suppkey", right_on="s_suppkey") \
        .merge(pdf["partsupp"], left_on=["l_partkey", "l_suppkey"],
               right_on=["ps_partkey", "ps_suppkey"]) \
        .merge(pdf["orders"], left_on="l_orderkey", right_on="o_orderkey") \
        .merge(pdf["nation"], left_on="s_nationkey", right_on="n_nationkey")
    j["o_year"] = pd.to_datetime(j.o_orderdate).dt.year.astype("int32")
    j["profit"] = _rev(j) - j.ps_supplycost * j.l_quantity
    g = j.groupby(["n_name", "o_year"], as_index=False).agg(
        sum_profit=("profit", "sum"))
    g = g.rename(columns={"n_name": "nation"})
    g = g.sort_values(["nation", "o_year"], ascending=[True, False]) \
        .reset_index(drop=True)
    got = run(9, tables)
    assert_frames(got, g)


def test_q10(tables, pdf):
    orders = pdf["orders"]
    orders = orders[(orders.o_orderdate >= dt.date(1993, 10, 1)) &
                    (orders.o_orderdate < dt.date(1994, 1, 1))]
    li = pdf["lineitem"][pdf["lineitem"].l_returnflag == "R"]
    j = li.merge(orders, left_on="l_orderkey", right_on="o_orderkey") \
        .merge(pdf["customer"], left_on="o_custkey", right_on="c_custkey") \
        .merge(pdf["nation"], left_on="c_nationkey", right_on="n_nationkey")
    j["revenue"] = _rev(j)
    g = j.groupby(["o_custkey", "c_name", "c_acctbal", "c_phone", "n_name",
                   "c_address", "c_comment"], as_index=False) \
        .agg(revenue=("revenue", "sum"))
    g = g.rename(columns={"o_custkey": "c_custkey"})[
        ["c_custkey", "c_name", "revenue", "c_acctbal", "n_name",
         "c_address", "c_phone", "c_comment"]]
    g = g.sort_values("revenue", ascending=False).head(20) \
        .reset_index(drop=True)
    got = run(10, tables)
    assert_frames(got, g, sort_cols=["revenue", "c_custkey"])


def test_q11(tables, pdf):
    nat = pdf["nation"][pdf["nation"].n_name == "GERMANY"]
    j = pdf["partsupp"] \
        .merge(pdf["supplier"], left_on="ps_suppkey", right_on="s_suppkey") \
        .merge(nat, left_on="s_nationkey", right_on="n_nationkey")
    j["value"] = j.ps_supplycost * j.ps_availqty
    g = j.groupby("ps_partkey", as_index=False).agg(value=("value", "sum"))
    thr = g.value.sum() * 0.0001 / SF
    g = g[g.value > thr].sort_values("value", ascending=False) \
        .reset_index(drop=True)
    got = run(11, tables)
    assert_frames(got, g, sort_cols=["value", "ps_partkey"])


def test_q12(tables, pdf):
    li = pdf["lineitem"]
    f = li[li.l_shipmode.isin(["MAIL", "SHIP"]) &
           (li.l_commitdate < li.l_receiptdate) &
           (li.l_shipdate < li.l_commitdate) &
           (li.l_receiptdate >= dt.date(1994, 1, 1)) &
           (li.l_receiptdate < dt.date(1995, 1, 1))]
    j = f.merge(pdf["orders"], left_on="l_orderkey", right_on="o_orderkey")
    hi = j.o_orderpriority.isin(["1-URGENT", "2-HIGH"])
    j["high"] = np.where(hi, 1, 0)
    j["low"] = np.where(hi, 0, 1)
    g = j.groupby("l_shipmode", as_index=False).agg(
        high_line_count=("high", "sum"), low_line_count=("low", "sum")) \
        .sort_values("l_shipmode").reset_index(drop=True)
    got = run(12, tables)
    assert_frames(got, g)


def test_q13(tables, pdf):
    orders = pdf["orders"]
    orders = orders[~orders.o_comment.str.match(".*special.*requests.*")]
    j = pdf["customer"].merge(orders, left_on="c_custkey",
                              right_on="o_custkey", how="left")
    g = j.groupby("c_custkey", as_index=False).agg(
        c_count=("o_orderkey", "count"))
    g2 = g.groupby("c_count", as_index=False).agg(
        custdist=("c_count", "count"))
    g2 = g2.sort_values(["custdist", "c_count"], ascending=[False, False]) \
        .reset_index(drop=True)
    got = run(13, tables)
    got["c_count"] = got["c_count"].astype("int64")
    g2["c_count"] = g2["c_count"].astype("int64")
    g2["custdist"] = g2["custdist"].astype("uint64")
    assert_frames(got, g2, sort_cols=["custdist", "c_count"])


def test_q14(tables, pdf):
    li = pdf["lineitem"]
    f = li[(li.l_shipdate >= dt.date(1995, 9, 1)) &
           (li.l_shipdate < dt.date(1995, 10, 1))]
    j = f.merge(pdf["part"], left_on="l_partkey", right_on="p_partkey")
    rev = _rev(j)
    promo = rev[j.p_type.str.startswith("PROMO")].sum()
    want = 100.0 * promo / rev.sum()
    got = run(14, tables)
    np.testing.assert_allclose(got["promo_revenue"][0], want, rtol=1e-9)


def test_q15(tables, pdf):
    li = pdf["lineitem"]
    f = li[(li.l_shipdate >= dt.date(1996, 1, 1)) &
           (li.l_shipdate < dt.date(1996, 4, 1))].copy()
    f["rev"] = _rev(f)
    g = f.groupby("l_suppkey", as_index=False).agg(
        total_revenue=("rev", "sum"))
    top = g.total_revenue.max()
    sel = g[g.total_revenue >= top - 1e-9]
    out = pdf["supplier"].merge(sel, left_on="s_suppkey",
                                right_on="l_suppkey")
    out = out[["s_suppkey", "s_name", "s_address", "s_phone",
               "total_revenue"]].sort_values("s_suppkey") \
        .reset_index(drop=True)
    got = run(15, tables)
    assert_frames(got, out)


def test_q16(tables, pdf):
    part = pdf["part"]
    part = part[(part.p_brand != "Brand#45") &
                ~part.p_type.str.startswith("MEDIUM POLISHED") &
                part.p_size.isin([49, 14, 23, 45, 19, 3, 36, 9])]
    bad = pdf["supplier"][pdf["supplier"].s_comment.str.match(
        ".*Customer.*Complaints.*")]
    ps = pdf["partsupp"].merge(part, left_on="ps_partkey",
                               right_on="p_partkey")
    ps = ps[~ps.ps_suppkey.isin(bad.s_suppkey)]
    g = ps.groupby(["p_brand", "p_type", "p_size"], as_index=False).agg(
        supplier_cnt=("ps_suppkey", "nunique"))
    g = g.sort_values(["supplier_cnt", "p_brand", "p_type", "p_size"],
                      ascending=[False, True, True, True]) \
        .reset_index(drop=True)
    g["supplier_cnt"] = g["supplier_cnt"].astype("uint64")
    got = run(16, tables)
    assert_frames(got, g, sort_cols=["supplier_cnt", "p_brand", "p_type",
                                     "p_size"])


def test_q17(tables, pdf):
    part = pdf["part"][(pdf["part"].p_brand == "Brand#23") &
                       (pdf["part"].p_container == "MED BOX")]
    j = pdf["lineitem"].merge(part, left_on="l_partkey",
                              right_on="p_partkey")
    avgs = j.groupby("l_partkey", as_index=False).agg(
        a=("l_quantity", "mean"))
    avgs["qty_limit"] = 0.2 * avgs.a
    jj = j.merge(avgs, on="l_partkey")
    sel = jj[jj.l_quantity < jj.qty_limit]
    want = sel.l_extendedprice.sum() / 7.0
    got = run(17, tables)
    if len(sel) == 0:
        assert got["avg_yearly"][0] is None or got["avg_yearly"].isna()[0]
    else:
        np.testing.assert_allclose(got["avg_yearly"][0], want, rtol=1e-9)


def test_q18(tables, pdf):
    li = pdf["lineitem"]
    sums = li.groupby("l_orderkey", as_index=False).agg(
        sum_qty=("l_quantity", "sum"))
    big = sums[sums.sum_qty > 300]
    j = pdf["orders"].merge(big, left_on="o_orderkey",
                            right_on="l_orderkey") \
        .merge(pdf["customer"], left_on="o_custkey", right_on="c_custkey")
    out = j[["c_name", "c_custkey", "o_orderkey", "o_orderdate",
             "o_totalprice", "sum_qty"]] \
        .sort_values(["o_totalprice", "o_orderdate"],
                     ascending=[False, True]).head(100) \
        .reset_index(drop=True)
    got = run(18, tables)
    assert_frames(got, out, sort_cols=["o_totalprice", "o_orderkey"])


def test_q19(tables, pdf):
    j = pdf["lineitem"].merge(pdf["part"], left_on="l_partkey",
                              right_on="p_partkey")
    sm = (j.p_brand == "Brand#12") & \
        j.p_container.isin(["SM CASE", "SM BOX", "SM PACK", "SM PKG"]) & \
        (j.l_quantity >= 1) & (j.l_quantity <= 11) & \
        (j.p_size >= 1) & (j.p_size <= 5)
    med = (j.p_brand == "Brand#23") & \
        j.p_container.isin(["MED BAG", "MED BOX", "MED PKG", "MED PACK"]) & \
        (j.l_quantity >= 10) & (j.l_quantity <= 20) & \
        (j.p_size >= 1) & (j.p_size <= 10)
    lg = (j.p_brand == "Brand#34") & \
        j.p_container.isin(["LG CASE", "LG BOX", "LG PACK", "LG PKG"]) & \
        (j.l_quantity >= 20) & (j.l_quantity <= 30) & \
        (j.p_size >= 1) & (j.p_size <= 15)
    common = j.l_shipmode.isin(["AIR", "AIR REG"]) & \
        (j.l_shipinstruct == "DELIVER IN PERSON")
    sel = j[common & (sm | med | lg)]
    want = _rev(sel).sum()
    got = run(19, tables)
    if len(sel) == 0:
        assert got["revenue"][0] is None or got["revenue"].isna()[0]
    else:
        np.testing.assert_allclose(got["revenue"][0], want, rtol=1e-9)


def test_q20(tables, pdf):
    part = pdf["part"][pdf["part"].p_name.str.startswith("forest")]
    li = pdf["lineitem"]
    li94 = li[(li.l_shipdate >= dt.date(1994, 1, 1)) &
              (li.l_shipdate < dt.date(1995, 1, 1))]
    qty = li94.groupby(["l_partkey", "l_suppkey"], as_index=False).agg(
        q=("l_quantity", "sum"))
    qty["half"] = 0.5 * qty.q
    ps = pdf["partsupp"]
    ps = ps[ps.ps_partkey.isin(part.p_partkey)]
    ps = ps.merge(qty, left_on=["ps_partkey", "ps_suppkey"],
                  right_on=["l_partkey", "l_suppkey"])
    ps = ps[ps.ps_availqty > ps.half]
    nat = pdf["nation"][pdf["nation"].n_name == "CANADA"]
    supp = pdf["supplier"].merge(nat, left_on="s_nationkey",
                                 right_on="n_nationkey")
    out = supp[supp.s_suppkey.isin(ps.ps_suppkey)][["s_name", "s_address"]] \
        .sort_values("s_name").reset_index(drop=True)
    got = run(20, tables)
    assert_frames(got, out)


def test_q21(tables, pdf):
    orders_f = pdf["orders"][pdf["orders"].o_orderstatus == "F"]
    li = pdf["lineitem"]
    li = li[li.l_orderkey.isin(orders_f.o_orderkey)]
    n_supp = li.groupby("l_orderkey").l_suppkey.nunique()
    late = li[li.l_receiptdate > li.l_commitdate]
    n_late = late.groupby("l_orderkey").l_suppkey.nunique()
    ok_orders = set(n_supp[n_supp > 1].index) & set(n_late[n_late == 1].index)
    sel = late[late.l_orderkey.isin(ok_orders)]
    nat = pdf["nation"][pdf["nation"].n_name == "SAUDI ARABIA"]
    supp = pdf["supplier"].merge(nat, left_on="s_nationkey",
                                 right_on="n_nationkey")
    j = sel.merge(supp, left_on="l_suppkey", right_on="s_suppkey")
    g = j.groupby("s_name", as_index=False).agg(numwait=("s_name", "count"))
    g = g.sort_values(["numwait", "s_name"], ascending=[False, True]) \
        .head(100).reset_index(drop=True)
    g["numwait"] = g["numwait"].astype("uint64")
    got = run(21, tables)
    assert_frames(got, g, sort_cols=["numwait", "s_name"])


def test_q22(tables, pdf):
    codes = ["13", "31", "23", "29", "30", "18", "17"]
    cust = pdf["customer"].copy()
    cust["cntrycode"] = cust.c_phone.str[:2]
    cust = cust[cust.cntrycode.isin(codes)]
    avg_bal = cust[cust.c_acctbal > 0.0].c_acctbal.mean()
    sel = cust[(cust.c_acctbal > avg_bal) &
               ~cust.c_custkey.isin(pdf["orders"].o_custkey)]
    g = sel.groupby("cntrycode", as_index=False).agg(
        numcust=("c_acctbal", "count"), totacctbal=("c_acctbal", "sum")) \
        .sort_values("cntrycode").reset_index(drop=True)
    g["numcust"] = g["numcust"].astype("uint64")
    got = run(22, tables)
    assert_frames(got, g)
