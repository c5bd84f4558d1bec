"""Series-level kernel tests (ref test pattern: tests/series/ in the
reference)."""
import datetime as dt

import pytest
import torch

from daft_amd import DataType, Series
from daft_amd.series import full_null


def test_from_pylist_roundtrip_ints():
    s = Series.from_pylist("a", [1, 2, None, 4])
    assert s.dtype == DataType.int64()
    assert s.to_pylist() == [1, 2, None, 4]
    assert s.null_count() == 1


def test_from_pylist_strings():
    s = Series.from_pylist("s", ["ab", None, "", "héllo"])
    assert s.dtype == DataType.string()
    assert s.to_pylist() == ["ab", None, "", "héllo"]


def test_from_pylist_dates():
    d = dt.date(2024, 3, 1)
    s = Series.from_pylist("d", [d, None])
    assert s.dtype == DataType.date()
    assert s.to_pylist() == [d, None]


def test_take_with_nulls():
    s = Series.from_pylist("a", [10, 20, 30])
    out = s.take(torch.tensor([2, -1, 0], dtype=torch.int64))
    assert out.to_pylist() == [30, None, 10]


def test_take_strings():
    s = Series.from_pylist("s", ["aa", "b", "ccc"])
    out = s.take(torch.tensor([2, 0, 2], dtype=torch.int64))
    assert out.to_pylist() == ["ccc", "aa", "ccc"]


def test_filter():
    s = Series.from_pylist("a", [1, 2, 3, 4])
    m = Series.from_pylist("m", [True, False, True, False])
    assert s.filter(m).to_pylist() == [1, 3]


def test_concat():
    a = Series.from_pylist("a", [1, 2])
    b = Series.from_pylist("a", [3, None])
    out = Series.concat([a, b])
    assert out.to_pylist() == [1, 2, 3, None]


def test_concat_strings():
    a = Series.from_pylist("a", ["x"])
    b = Series.from_pylist("a", ["yy", "z"])
    assert Series.concat([a, b]).to_pylist() == ["x", "yy", "z"]


def test_binary_ops_nulls():
    a = Series.from_pylist("a", [1, None, 3])
    b = Series.from_pylist("b", [10, 20, None])
    assert (a + b).to_pylist() == [11, None, None]
    assert (a * b).to_pylist() == [10, None, None]


def test_division_promotes_float():
    a = Series.from_pylist("a", [1, 2])
    b = Series.from_pylist("b", [2, 4])
    out = a / b
    assert out.dtype == DataType.float64()
    assert out.to_pylist() == [0.5, 0.5]


def test_compare():
    a = Series.from_pylist("a", [1, 2, 3])
    b = Series.from_pylist("b", [2, 2, 2])
    assert a.compare(b, "lt").to_pylist() == [True, False, False]
    assert a.compare(b, "eq").to_pylist() == [False, True, False]


def test_string_compare():
    a = Series.from_pylist("a", ["a", "b", None])
    b = Series.from_pylist("b", ["a", "a", "a"])
    out = a.compare(b, "eq")
    assert out.to_pylist() == [True, False, None]


def test_logical_three_valued():
    a = Series.from_pylist("a", [True, True, False, None])
    b = Series.from_pylist("b", [True, None, None, None])
    assert a.logical(b, "and").to_pylist() == [True, None, False, None]
    assert a.logical(b, "or").to_pylist() == [True, True, None, None]


def test_cast():
    s = Series.from_pylist("a", [1, 2, 3])
    assert s.cast(DataType.float32()).to_pylist() == [1.0, 2.0, 3.0]
    assert s.cast(DataType.string()).to_pylist() == ["1", "2", "3"]


def test_is_in():
    s = Series.from_pylist("a", [1, 2, 3, 4])
    vals = Series.from_pylist("v", [2, 4])
    assert s.is_in(vals).to_pylist() == [False, True, False, True]


def test_if_else():
    c = Series.from_pylist("c", [True, False, True])
    t = Series.from_pylist("t", [1, 2, 3])
    f = Series.from_pylist("f", [10, 20, 30])
    assert c.if_else(t, f).to_pylist() == [1, 20, 3]


def test_full_null():
    s = full_null("x", DataType.string(), 3)
    assert s.to_pylist() == [None, None, None]


def test_list_series():
    s = Series.from_pylist("l", [[1, 2], [], None, [3]])
    assert s.dtype == DataType.list(DataType.int64())
    assert s.to_pylist() == [[1, 2], [], None, [3]]
    out = s.take(torch.tensor([3, 0], dtype=torch.int64))
    assert out.to_pylist() == [[3], [1, 2]]


def test_struct_series():
    s = Series.from_pylist("st", [{"x": 1, "y": "a"}, {"x": 2, "y": "b"}])
    assert s.to_pylist() == [{"x": 1, "y": "a"}, {"x": 2, "y": "b"}]


def test_embedding_series():
    s = Series.from_pylist("e", [[1.0, 2.0], [3.0, 4.0]],
                           DataType.embedding(DataType.float32(), 2))
    assert s.to_pylist() == [[1.0, 2.0], [3.0, 4.0]]


def test_arrow_roundtrip():
    import pyarrow as pa
    s = Series.from_pylist("a", [1, None, 3])
    arr = s.to_arrow()
    assert arr.to_pylist() == [1, None, 3]
    s2 = Series.from_arrow("a", arr)
    assert s2.to_pylist() == [1, None, 3]


def test_arrow_roundtrip_strings():
    s = Series.from_pylist("s", ["ab", None, "cd"])
    arr = s.to_arrow()
    assert arr.to_pylist() == ["ab", None, "cd"]
    assert Series.from_arrow("s", arr).to_pylist() == ["ab", None, "cd"]


def test_hash_consistency():
    a = Series.from_pylist("a", [1, 2, 1, None, None])
    h = a.hash()
    assert h[0] == h[2]
    assert h[3] == h[4]
    assert h[0] != h[1]


def test_argsort_nulls_last():
    s = Series.from_pylist("a", [3, None, 1, 2])
    perm = s.argsort()
    assert s.take(perm).to_pylist() == [1, 2, 3, None]


def test_argsort_desc():
    s = Series.from_pylist("a", [3, None, 1, 2])
    perm = s.argsort(descending=True, nulls_first=True)
    assert s.take(perm).to_pylist() == [None, 3, 2, 1]


def test_decimal_exact_storage_and_ops(tmp_path):
    """Decimal128 (p<=18) stores scaled int64: sums/joins/sorts are exact
    (ref: daft-core Decimal128Array semantics)."""
    from decimal import Decimal as D
    import daft_amd as daft
    from daft_amd import col

    df = daft.from_pydict({"v": [D("0.10")] * 100 + [D("0.05")]})
    tot = df.agg(col("v").sum().alias("s")).to_pydict()["s"][0]
    assert tot == D("10.05")  # float64 would drift

    # arithmetic scale rules
    d2 = daft.from_pydict({"a": [D("1.25")], "b": [D("0.4")]})
    out = d2.select((col("a") + col("b")).alias("s"),
                    (col("a") * col("b")).alias("m"),
                    (col("a") - D("0.05")).alias("d")).to_pydict()
    assert out["s"][0] == D("1.65")
    assert out["m"][0] == D("0.500")
    assert out["d"][0] == D("1.20")

    # join + groupby on decimal keys are exact
    l = daft.from_pydict({"k": [D("2.50"), D("1.10"), D("2.50")]})
    r = daft.from_pydict({"k": [D("1.10"), D("2.50")], "w": [1, 2]})
    j = l.join(r, on="k").groupby("k").agg(col("w").sum().alias("t")) \
        .sort("k").to_pydict()
    assert j["t"] == [1, 4]

    # parquet round trip preserves exact values
    import pyarrow.parquet as pq
    p = str(tmp_path / "d.parquet")
    daft.from_pydict({"v": [D("123.45"), None, D("-0.01")]}) \
        .write_parquet(p)
    import glob
    files = glob.glob(p + "/*.parquet") if not p.endswith(".parquet") or \
        __import__("os").path.isdir(p) else [p]
    back = daft.read_parquet(files if files else p).to_pydict()["v"]
    assert back == [D("123.45"), None, D("-0.01")]

    # cast decimal -> string keeps trailing zeros per scale
    s = daft.from_pydict({"v": [D("1.10")]}).select(
        col("v").cast(daft.DataType.string()).alias("s")).to_pydict()["s"]
    assert s == ["1.10"]


def test_is_in_sorted_table_path():
    import random
    random.seed(9)
    vals = [random.randint(0, 1000) if i % 11 else None
            for i in range(5000)]
    members = [3, 77, 500, 999, 123, 456, 789]   # >4 -> searchsorted path
    s = Series.from_pylist("x", vals, DataType.int64())
    m = Series.from_pylist("v", members, DataType.int64())
    got = s.is_in(m).to_pylist()
    want = [None if v is None else (v in set(members)) for v in vals]
    assert got == want
    # floats too
    fvals = [float(v) if v is not None else None for v in vals]
    sf = Series.from_pylist("x", fvals, DataType.float64())
    mf = Series.from_pylist("v", [float(x) for x in members],
                            DataType.float64())
    assert sf.is_in(mf).to_pylist() == want


def test_map_first_class():
    """First-class Map type: construction from dicts, structural ops,
    map_get/keys/values, and arrow round trip (ref: daft-schema Map ->
    List(Struct) to_physical; daft-sql map module)."""
    import torch
    import daft_amd as daft
    from daft_amd import col
    from daft_amd import arrow_interop as ai
    dt = DataType.map(DataType.string(), DataType.int64())
    s = Series.from_pylist("m", [{"a": 1}, {"b": 2, "c": 3}, None,
                                 {"d": 4}], dt)
    assert s.to_pylist() == [{"a": 1}, {"b": 2, "c": 3}, None, {"d": 4}]
    assert s.take(torch.tensor([3, 1])).to_pylist() == \
        [{"d": 4}, {"b": 2, "c": 3}]
    assert s.slice(1, 3).to_pylist() == [{"b": 2, "c": 3}, None]
    assert Series.concat([s, s]).to_pylist()[5] == {"b": 2, "c": 3}
    # arrow round trip through pa.map_
    arr = ai.to_arrow_array(s)
    import pyarrow as pa
    assert pa.types.is_map(arr.type)
    back = ai.from_arrow_array("m", arr)
    assert back.to_pylist() == s.to_pylist()
    # functions
    from daft_amd.functions import map_get, map_keys
    from daft_amd.functions.aliases import map_values
    df = daft.from_pydict({"m": s})
    out = df.select(map_get(col("m"), "b").alias("b"),
                    map_keys(col("m")).alias("k"),
                    map_values(col("m")).alias("v")).to_pydict()
    assert out["b"] == [None, 2, None, None]
    assert out["k"] == [["a"], ["b", "c"], None, ["d"]]
    assert out["v"] == [[1], [2, 3], None, [4]]
