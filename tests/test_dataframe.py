"""DataFrame integration tests (ref pattern: tests/dataframe/ in the
reference — joins, aggs, distinct, explode, sample, sort...)."""
import datetime as dt

import pytest

import daft_amd as daft
from daft_amd import DataType, col, lit
from tests.conftest import assert_df_eq


@pytest.fixture
def df():
    return daft.from_pydict({
        "a": [1, 2, 3, 4, 5, 6],
        "b": ["x", "y", "x", "z", "y", "x"],
        "c": [1.5, None, 2.5, 3.0, 4.5, -1.0],
    })


def test_select_project(df):
    out = df.select(col("a"), (col("a") + 10).alias("a10")).to_pydict()
    assert out == {"a": [1, 2, 3, 4, 5, 6], "a10": [11, 12, 13, 14, 15, 16]}


def test_filter_chain(df):
    out = df.where(col("a") > 2).where(col("b") == "x").to_pydict()
    assert out["a"] == [3, 6]


def test_with_column(df):
    out = df.with_column("d", col("a") * col("a")).to_pydict()
    assert out["d"] == [1, 4, 9, 16, 25, 36]


def test_limit_offset(df):
    assert df.limit(2).to_pydict()["a"] == [1, 2]
    assert df.offset(4).to_pydict()["a"] == [5, 6]


def test_sort_multi(df):
    out = df.sort(["b", "a"], desc=[False, True]).to_pydict()
    assert out["b"] == ["x", "x", "x", "y", "y", "z"]
    assert out["a"] == [6, 3, 1, 5, 2, 4]


def test_distinct():
    df = daft.from_pydict({"a": [1, 1, 2, 2, 3], "b": [1, 1, 2, 9, 3]})
    out = df.distinct().sort(["a", "b"]).to_pydict()
    assert out == {"a": [1, 2, 2, 3], "b": [1, 2, 9, 3]}


def test_groupby_aggs(df):
    out = df.groupby("b").agg(
        col("a").sum().alias("sa"),
        col("c").mean().alias("mc"),
        col("a").count().alias("cnt"),
        col("c").min().alias("mn"),
        col("c").max().alias("mx"),
    ).sort("b").to_pydict()
    assert out["b"] == ["x", "y", "z"]
    assert out["sa"] == [10, 7, 4]
    assert out["cnt"] == [3, 2, 1]
    assert out["mn"] == [-1.0, 4.5, 3.0]
    assert out["mx"] == [2.5, 4.5, 3.0]
    assert out["mc"] == pytest.approx([1.0, 4.5, 3.0])


def test_global_agg(df):
    out = df.agg(col("a").sum().alias("s"),
                 col("c").count().alias("n")).to_pydict()
    assert out == {"s": [21], "n": [5]}


def test_agg_compound_expr(df):
    out = df.groupby("b").agg(
        (col("a").sum() * 2 + col("a").count()).alias("w")
    ).sort("b").to_pydict()
    assert out["w"] == [23, 16, 9]


def test_count_distinct():
    df = daft.from_pydict({"g": [1, 1, 1, 2, 2], "v": [1, 1, 2, 5, None]})
    out = df.groupby("g").agg(
        col("v").count_distinct().alias("nd")).sort("g").to_pydict()
    assert out["nd"] == [2, 1]


def test_any_value_and_list():
    df = daft.from_pydict({"g": [1, 1, 2], "v": [10, 20, 30]})
    out = df.groupby("g").agg(
        col("v").any_value().alias("av"),
        col("v").agg_list().alias("lst")).sort("g").to_pydict()
    assert out["av"] == [10, 30]
    assert out["lst"] == [[10, 20], [30]]


def test_stddev():
    df = daft.from_pydict({"v": [1.0, 2.0, 3.0, 4.0]})
    out = df.agg(col("v").stddev().alias("sd"),
                 col("v").variance().alias("var")).to_pydict()
    assert out["var"][0] == pytest.approx(1.25)
    assert out["sd"][0] == pytest.approx(1.25 ** 0.5)


@pytest.mark.parametrize("how,expected_a", [
    ("inner", [1, 2, 3]),
    ("left", [1, 2, 3, 4]),
    ("semi", [1, 2, 3]),
    ("anti", [4]),
])
def test_join_types(how, expected_a):
    l = daft.from_pydict({"k": [1, 2, 3, 4], "a": [1, 2, 3, 4]})
    r = daft.from_pydict({"k": [1, 2, 3, 3], "v": [10, 20, 30, 31]})
    out = l.join(r.distinct("k"), on="k", how=how).sort("a").to_pydict()
    assert out["a"] == expected_a


def test_join_outer():
    l = daft.from_pydict({"k": [1, 2], "a": [1, 2]})
    r = daft.from_pydict({"k": [2, 3], "v": [20, 30]})
    out = l.join(r, on="k", how="outer").sort("k").to_pydict()
    assert out["k"] == [1, 2, 3]
    assert out["a"] == [1, 2, None]
    assert out["v"] == [None, 20, 30]


def test_join_duplicate_names_suffix():
    l = daft.from_pydict({"k": [1], "v": [1]})
    r = daft.from_pydict({"k": [1], "v": [2]})
    out = l.join(r, on="k").to_pydict()
    assert out == {"k": [1], "v": [1], "v_right": [2]}


def test_join_null_keys_dont_match():
    l = daft.from_pydict({"k": [1, None], "a": [1, 2]})
    r = daft.from_pydict({"k": [1, None], "v": [10, 20]})
    out = l.join(r, on="k", how="inner").to_pydict()
    assert out["a"] == [1]


def test_cross_join():
    l = daft.from_pydict({"a": [1, 2]})
    r = daft.from_pydict({"b": ["x", "y", "z"]})
    out = l.join(r, how="cross").to_pydict()
    assert len(out["a"]) == 6


def test_string_key_join():
    l = daft.from_pydict({"k": ["aa", "bb", "cc"], "a": [1, 2, 3]})
    r = daft.from_pydict({"k": ["bb", "cc", "dd"], "v": [20, 30, 40]})
    out = l.join(r, on="k").sort("a").to_pydict()
    assert out["k"] == ["bb", "cc"]
    assert out["v"] == [20, 30]


def test_concat():
    a = daft.from_pydict({"x": [1, 2]})
    b = daft.from_pydict({"x": [3]})
    assert a.concat(b).to_pydict()["x"] == [1, 2, 3]


def test_explode():
    df = daft.from_pydict({"a": [1, 2, 3], "l": [[10, 20], [], [30]]})
    out = df.explode("l").to_pydict()
    assert out["a"] == [1, 1, 2, 3]
    assert out["l"] == [10, 20, None, 30]


def test_unpivot():
    df = daft.from_pydict({"id": [1, 2], "x": [10, 20], "y": [30, 40]})
    out = df.unpivot(["id"], ["x", "y"]).sort(["id", "variable"]).to_pydict()
    assert out["id"] == [1, 1, 2, 2]
    assert out["variable"] == ["x", "x", "y", "y"][0:1] + ["y", "x", "y"] \
        if False else out["variable"] == ["x", "y", "x", "y"]
    assert out["value"] == [10, 30, 20, 40]


def test_pivot():
    df = daft.from_pydict({"g": [1, 1, 2], "p": ["a", "b", "a"],
                           "v": [10, 20, 30]})
    out = df.pivot("g", col("p"), col("v"), "sum").sort("g").to_pydict()
    assert out["g"] == [1, 2]
    assert out["a"] == [10, 30]
    assert out["b"] == [20, None]


def test_sample_fraction(df):
    out = df.sample(0.5, seed=42).to_pydict()
    assert 0 <= len(out["a"]) <= 6


def test_monotonic_id(df):
    out = df.add_monotonically_increasing_id("id").to_pydict()
    assert out["id"] == [0, 1, 2, 3, 4, 5]


def test_if_else_expr(df):
    out = df.select(
        (col("a") > 3).if_else(lit("big"), lit("small")).alias("sz")
    ).to_pydict()
    assert out["sz"] == ["small", "small", "small", "big", "big", "big"]


def test_is_in(df):
    out = df.where(col("b").is_in(["x", "z"])).to_pydict()
    assert out["a"] == [1, 3, 4, 6]


def test_between(df):
    out = df.where(col("a").between(2, 4)).to_pydict()
    assert out["a"] == [2, 3, 4]


def test_fill_null(df):
    out = df.select(col("c").fill_null(0.0)).to_pydict()
    assert out["c"] == [1.5, 0.0, 2.5, 3.0, 4.5, -1.0]


def test_coalesce(df):
    from daft_amd.functions import coalesce
    out = df.select(coalesce(col("c"), col("a") * 1.0).alias("cc")).to_pydict()
    assert out["cc"] == [1.5, 2.0, 2.5, 3.0, 4.5, -1.0]


def test_repartition_roundtrip(df):
    out = df.repartition(3, "b").sort("a").to_pydict()
    assert out["a"] == [1, 2, 3, 4, 5, 6]


def test_into_partitions(df):
    d2 = df.into_partitions(3).collect()
    assert d2.num_partitions() >= 1
    assert sorted(d2.to_pydict()["a"]) == [1, 2, 3, 4, 5, 6]


def test_empty_result(df):
    out = df.where(col("a") > 100).to_pydict()
    assert out["a"] == []


def test_count_rows(df):
    assert df.count_rows() == 6
    assert df.where(col("b") == "x").count_rows() == 3


def test_union_intersect():
    a = daft.from_pydict({"x": [1, 2, 3]})
    b = daft.from_pydict({"x": [2, 3, 4]})
    assert sorted(a.union(b).to_pydict()["x"]) == [1, 2, 3, 4]
    assert sorted(a.intersect(b).to_pydict()["x"]) == [2, 3]
    assert a.except_distinct(b).to_pydict()["x"] == [1]


def test_iter_rows(df):
    rows = list(df.limit(2).iter_rows())
    assert rows[0] == {"a": 1, "b": "x", "c": 1.5}


def test_to_pandas(df):
    pdf = df.to_pandas()
    assert list(pdf.columns) == ["a", "b", "c"]
    assert len(pdf) == 6


def test_schema_and_getitem(df):
    assert df.schema["a"].dtype == DataType.int64()
    out = df.select(df["a"] + 1).to_pydict()
    assert out["a"] == [2, 3, 4, 5, 6, 7]


def test_topn_rewrite(df):
    out = df.sort("a", desc=True).limit(2).to_pydict()
    assert out["a"] == [6, 5]


def test_join_asof_backward():
    trades = daft.from_pydict({
        "t": [3, 7, 10], "sym": ["a", "a", "b"], "px": [1.0, 2.0, 3.0]})
    quotes = daft.from_pydict({
        "t": [1, 5, 8, 9], "sym": ["a", "a", "a", "b"],
        "bid": [10.0, 20.0, 30.0, 40.0]})
    out = trades.join_asof(quotes, left_on="t", right_on="t",
                           by=["sym"]).sort("t").to_pydict()
    assert out["bid"] == [10.0, 20.0, 40.0]


def test_join_asof_forward_no_by():
    l = daft.from_pydict({"t": [2, 6], "v": [1, 2]})
    r = daft.from_pydict({"t": [4, 5], "w": [40, 50]})
    out = l.join_asof(r, left_on="t", right_on="t",
                      strategy="forward").sort("t").to_pydict()
    assert out["w"] == [40, None]


def test_streamed_aggregate_and_join_multibatch(tmp_path):
    """Multiple input batches take the partial/final streaming path in
    AggregateOp and the per-batch probe path in JoinOp; results must match
    the single-batch plan."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    paths = []
    for i in range(4):
        p = str(tmp_path / f"f{i}.parquet")
        pq.write_table(pa.table({
            "g": [f"k{j % 5}" for j in range(50)],
            "v": [float(i * 50 + j) for j in range(50)],
            "k": [(i * 50 + j) % 17 for j in range(50)],
        }), p)
        paths.append(p)
    multi = daft.read_parquet(paths)
    single = daft.from_pydict(multi.to_pydict())

    for df_pair in [(multi, single)]:
        m, s = df_pair
        am = m.groupby("g").agg(
            col("v").sum().alias("s"), col("v").mean().alias("mu"),
            col("v").count().alias("c"), col("v").min().alias("lo"),
            col("v").stddev().alias("sd")).sort("g").to_pydict()
        as_ = s.groupby("g").agg(
            col("v").sum().alias("s"), col("v").mean().alias("mu"),
            col("v").count().alias("c"), col("v").min().alias("lo"),
            col("v").stddev().alias("sd")).sort("g").to_pydict()
        assert am["s"] == as_["s"] and am["c"] == as_["c"]
        assert am["mu"] == pytest.approx(as_["mu"])
        assert am["sd"] == pytest.approx(as_["sd"])

    # ungrouped
    assert multi.agg(col("v").sum().alias("s")).to_pydict() == \
        single.agg(col("v").sum().alias("s")).to_pydict()

    # non-decomposable agg falls back to materialize
    lm = multi.groupby("g").agg(col("v").agg_list().alias("l")) \
        .sort("g").to_pydict()
    ls = single.groupby("g").agg(col("v").agg_list().alias("l")) \
        .sort("g").to_pydict()
    assert [sorted(x) for x in lm["l"]] == [sorted(x) for x in ls["l"]]

    # streamed probe join
    dim = daft.from_pydict({"k": list(range(17)),
                            "name": [f"n{i}" for i in range(17)]})
    jm = multi.join(dim, on="k").sort("v").to_pydict()
    js = single.join(dim, on="k").sort("v").to_pydict()
    assert jm == js
    sm = multi.join(dim.where(col("k") < 5), on="k", how="semi") \
        .sort("v").to_pydict()
    ss = single.join(dim.where(col("k") < 5), on="k", how="semi") \
        .sort("v").to_pydict()
    assert sm == ss


def test_dense_key_join_matches_hash_join():
    """Direct-address PK-join fast path vs the hash join (rowops)."""
    import numpy as np
    from daft_amd.series import Series
    from daft_amd.schema import DataType
    from daft_amd.kernels import rowops
    rng = np.random.default_rng(11)
    for how in ("inner", "left", "right", "outer", "semi", "anti"):
        rk = [int(v) for v in rng.permutation(np.arange(100, 300))[:150]]
        lvals = [None if rng.random() < 0.1 else int(v)
                 for v in rng.integers(50, 350, 400)]
        lk = [Series.from_pylist("k", lvals, DataType.int64())]
        rkS = [Series.from_pylist("k", rk, DataType.int64())]
        d = rowops._dense_key_join(lk, rkS, how)
        assert d is not None, how
        li1, ri1 = d
        li2, ri2 = rowops._cpu_join(lk, rkS, how)
        if how in ("semi", "anti"):
            assert sorted(li1.tolist()) == sorted(li2.tolist()), how
        else:
            assert sorted(zip(li1.tolist(), ri1.tolist())) == \
                sorted(zip(li2.tolist(), ri2.tolist())), how
    # duplicate build keys: semi/anti stay on the dense path (existence
    # only); fan-out joins fall back to the hash join
    dupr = [Series.from_pylist("k", [1, 2, 2, 7], DataType.int64())]
    for how in ("semi", "anti"):
        d = rowops._dense_key_join(lk, dupr, how)
        assert d is not None
        li1, _ = d
        li2, _ = rowops._cpu_join(lk, dupr, how)
        assert sorted(li1.tolist()) == sorted(li2.tolist()), how
    assert rowops._dense_key_join(
        lk, [Series.from_pylist("k", [1, 2, 2], DataType.int64())],
        "inner") is None
    assert rowops._dense_key_join(
        lk, [Series.from_pylist("k", [1, 10**9], DataType.int64())],
        "inner") is None


def test_dataframe_parity_methods():
    """Reference-parity method batch: set ops, agg shortcuts, describe,
    drop_nan/null, map_groups, write_sink."""
    df = daft.from_pydict({"a": [1, 2, 2, None],
                           "b": [1.0, float("nan"), 3.0, 4.0]})
    assert df.columns == ["a", "b"]
    assert df.drop_null("a").count_rows() == 3
    assert df.drop_nan("b").count_rows() == 3
    d = df.describe().to_pydict()
    assert d["column"] == ["a", "b"] and d["nulls"] == [1, 0]
    u = daft.from_pydict({"a": [9], "c": ["x"]})
    ub = df.union_all_by_name(u).to_pydict()
    assert set(ub.keys()) == {"a", "b", "c"} and len(ub["a"]) == 5
    l = daft.from_pydict({"x": [1, 1, 2, 3]})
    r = daft.from_pydict({"x": [1, 2, 2]})
    assert sorted(l.intersect_all(r).to_pydict()["x"]) == [1, 2]
    assert sorted(l.except_all(r).to_pydict()["x"]) == [1, 3]
    assert df.count_distinct("a").to_pydict()["a"] == [2]
    g = daft.from_pydict({"k": ["a", "a", "b"], "v": [1, 2, 3]})
    mg = g.map_groups(lambda s: s.sum("v"), "k").to_pydict()
    assert sorted(mg["v"]) == [3, 3]

    class Sink:
        rows = 0

        def write(self, b):
            Sink.rows += len(b)
            return len(b)

        def finalize(self, rs):
            return sum(rs)
    df.write_sink(Sink())
    assert Sink.rows == 4
    with pytest.raises(RuntimeError):
        df.write_deltalake("x")


def test_packed_two_key_join_matches():
    import numpy as np
    rng = np.random.default_rng(5)
    n = 500
    l = daft.from_pydict({
        "a": [int(v) for v in rng.integers(0, 30, n)],
        "b": [int(v) for v in rng.integers(0, 10, n)],
    })
    r = daft.from_pydict({
        "a": [int(v) for v in rng.integers(0, 30, 120)],
        "b": [int(v) for v in rng.integers(0, 10, 120)],
        "w": list(range(120)),
    })
    got = l.join(r, on=["a", "b"]).sort(["a", "b", "w"]).to_pydict()
    # reference result via pandas merge
    import pandas as pd
    want = pd.merge(l.to_pandas(), r.to_pandas(), on=["a", "b"]) \
        .sort_values(["a", "b", "w"])
    assert got["w"] == want["w"].tolist()
    # with nulls on one side
    l2 = daft.from_pydict({"a": [1, None, 2], "b": [1, 1, None]})
    r2 = daft.from_pydict({"a": [1, 2], "b": [1, 3], "w": [10, 20]})
    out = l2.join(r2, on=["a", "b"]).to_pydict()
    assert out["w"] == [10]


def test_streaming_sinks_multibatch(tmp_path):
    """Sort / window / asof over multi-batch (multi-file) inputs match
    the single-batch plan (blocking sinks materialize correctly)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    paths = []
    for i in range(3):
        p = str(tmp_path / f"f{i}.parquet")
        pq.write_table(pa.table({
            "g": [f"k{j % 4}" for j in range(40)],
            "v": [float((i * 40 + j) % 23) for j in range(40)],
            "t": [i * 40 + j for j in range(40)],
        }), p)
        paths.append(p)
    multi = daft.read_parquet(paths)
    single = daft.from_pydict(multi.to_pydict())

    assert multi.sort(["v", "t"]).to_pydict() == \
        single.sort(["v", "t"]).to_pydict()

    from daft_amd.window import Window
    from daft_amd.functions import row_number
    w = Window().partition_by("g").order_by("t")
    wm = multi.with_window_columns({"rn": row_number().over(w)}) \
        .sort("t").to_pydict()
    ws = single.with_window_columns({"rn": row_number().over(w)}) \
        .sort("t").to_pydict()
    assert wm == ws

    right = daft.from_pydict({"t": [10, 50, 90], "mark": ["a", "b", "c"]})
    am = multi.join_asof(right, left_on="t", right_on="t") \
        .sort("t").to_pydict()
    asg = single.join_asof(right, left_on="t", right_on="t") \
        .sort("t").to_pydict()
    assert am == asg


def test_projection_cse_evaluates_shared_subtree_once():
    """Pure shared subexpressions in a projection evaluate once per
    batch (physical/cse.py); results are unchanged."""
    from daft_amd.expressions import expressions as E
    calls = {"n": 0}
    orig = E.BinaryOp.evaluate

    def spy(self, batch):
        if self.op == "add":
            calls["n"] += 1
        return orig(self, batch)
    E.BinaryOp.evaluate = spy
    try:
        df = daft.from_pydict({"a": [1, 2], "b": [10, 20]})
        out = df.select(((col("a") + col("b")) * 2).alias("x"),
                        ((col("a") + col("b")) - 1).alias("y")).to_pydict()
    finally:
        E.BinaryOp.evaluate = orig
    assert out == {"x": [22, 44], "y": [10, 21]}
    assert calls["n"] == 1


@pytest.mark.parametrize("nparts", [1, 2, 5])
def test_partition_sweep_invariance(nparts):
    """Results are invariant to the partition count (the reference's
    conftest partitioning sweep)."""
    data = {"g": [f"k{i % 4}" for i in range(97)],
            "v": [float(i % 13) for i in range(97)],
            "k": [i % 7 for i in range(97)]}
    base = daft.from_pydict(data)
    df = base.into_partitions(nparts) if nparts > 1 else base

    want_agg = base.groupby("g").agg(col("v").sum().alias("s")) \
        .sort("g").to_pydict()
    assert df.groupby("g").agg(col("v").sum().alias("s")) \
        .sort("g").to_pydict() == want_agg

    dim = daft.from_pydict({"k": list(range(7)),
                            "w": [i * 10 for i in range(7)]})
    want_j = base.join(dim, on="k").sort(["v", "k", "w"]).to_pydict()
    assert df.join(dim, on="k").sort(["v", "k", "w"]).to_pydict() == want_j

    assert df.sort("v").to_pydict()["v"] == \
        base.sort("v").to_pydict()["v"]
    assert df.distinct("g").count_rows() == 4
    assert df.count_rows() == 97


def test_empty_input_sweep_new_surfaces():
    """Zero-row inputs through the newer surfaces (describe, set ops,
    union_by_name, SQL set ops, agg_set)."""
    e = daft.from_pydict({"a": [], "g": []})
    assert e.describe().to_pydict()["count"] == [0, 0]
    assert e.drop_null("a").count_rows() == 0
    assert e.groupby("g").agg(col("a").sum().alias("s")).count_rows() == 0
    assert e.intersect_all(e).count_rows() == 0
    assert e.except_all(e).count_rows() == 0
    u = e.union_all_by_name(daft.from_pydict({"a": [1], "z": ["x"]}))
    assert u.count_rows() == 1
    assert daft.sql("select a from e union select a from e") \
        .to_pydict()["a"] == []
