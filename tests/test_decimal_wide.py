"""Exact wide Decimal128 (precision 19..38): two-int64-limb storage
(struct<lo,hi> physical), carry arithmetic, comparisons, sorting,
SUM/MIN/MAX aggregation, casts and arrow interop — all verified against
Python's arbitrary-precision Decimal/int oracle (ref capability:
/root/reference/src/daft-core/ Decimal128 logical type)."""
import decimal
import random

decimal.getcontext().prec = 80      # oracle must be exact, not 28-digit

import pytest
import torch

import daft_amd as daft
from daft_amd import col
from daft_amd.schema import DataType
from daft_amd.series import Series

D = decimal.Decimal
WIDE = DataType.decimal128(38, 10)


def _mk(vals, dtype=WIDE, name="d"):
    return Series.from_pylist(name, vals, dtype)


def _rand_wide(n, scale=10, digits=37, seed=0):
    rng = random.Random(seed)
    out = []
    for _ in range(n):
        mag = rng.randint(0, 10 ** digits - 1)
        v = D(mag) * (1 if rng.random() < 0.5 else -1)
        out.append(v.scaleb(-scale))
    return out


def test_roundtrip_extremes():
    vals = [D("12345678901234567890123456.7890123456"),
            D("-9999999999999999999999999999.9999999999"),
            None,
            D("0"),
            D("-0.0000000001"),
            D((10 ** 37 - 1)).scaleb(-10)]
    s = _mk(vals)
    assert s.dtype == WIDE
    assert s.children is not None and s.data is None   # limb storage
    assert s.to_pylist() == vals


def test_selection_ops():
    vals = _rand_wide(100)
    s = _mk(vals)
    idx = torch.tensor([5, 0, 99, 5, -1])
    got = s.take(idx).to_pylist()
    assert got == [vals[5], vals[0], vals[99], vals[5], None]
    assert s.slice(10, 13).to_pylist() == vals[10:13]
    both = Series.concat([s, s.slice(0, 2)])
    assert both.to_pylist() == vals + vals[:2]


@pytest.mark.parametrize("op", ["add", "sub", "mul"])
def test_exact_arithmetic(op):
    a = _rand_wide(200, scale=10, digits=17, seed=1)
    b = _rand_wide(200, scale=10, digits=17, seed=2)
    sa, sb = _mk(a), _mk(b)
    if op == "add":
        got, want = (sa + sb), [x + y for x, y in zip(a, b)]
    elif op == "sub":
        got, want = (sa - sb), [x - y for x, y in zip(a, b)]
    else:
        got, want = (sa * sb), [x * y for x, y in zip(a, b)]
    assert got.dtype.is_decimal() and got.dtype.precision > 18
    assert got.to_pylist() == want


def test_narrow_mul_promotes_to_wide_exact():
    # p18 x p18 overflows int64: the product must come back exact, wide
    a = [D("123456789012345.678"), D("-999999999999999.999")]
    b = [D("987654321098765.432"), D("999999999999999.999")]
    na = _mk(a, DataType.decimal128(18, 3))
    nb = _mk(b, DataType.decimal128(18, 3))
    got = na * nb
    assert got.dtype.precision > 18
    assert got.to_pylist() == [x * y for x, y in zip(a, b)]


def test_mixed_narrow_wide_add():
    w = [D("12345678901234567890123456.7890123456")]
    n = [D("0.01")]
    got = (_mk(w) + _mk(n, DataType.decimal128(4, 2))).to_pylist()
    assert got == [w[0] + n[0]]


def test_compare_and_if_else():
    a = _rand_wide(300, seed=3)
    b = _rand_wide(300, seed=4)
    sa, sb = _mk(a), _mk(b)
    from daft_amd.kernels import compare_op
    for op, fn in (("lt", lambda x, y: x < y), ("ge", lambda x, y: x >= y),
                   ("eq", lambda x, y: x == y),
                   ("gt", lambda x, y: x > y)):
        got = compare_op(sa, sb, op).to_pylist()
        assert got == [fn(x, y) for x, y in zip(a, b)]
    df = daft.from_pydict({"a": a, "b": b})
    out = df.select(
        (col("a") > col("b")).if_else(col("a"), col("b")).alias("mx")
    ).to_pydict()["mx"]
    assert out == [max(x, y) for x, y in zip(a, b)]


def test_sort_wide():
    vals = _rand_wide(500, seed=5) + [None, D("0.0000000001"), None]
    df = daft.from_pydict({"d": vals}).sort("d")
    got = df.to_pydict()["d"]
    nn = [v for v in got if v is not None]
    assert nn == sorted(v for v in vals if v is not None)


def test_grouped_sum_min_max_exact():
    rng = random.Random(6)
    keys, vals = [], []
    for i in range(1000):
        keys.append(rng.randint(0, 7))
        mag = rng.randint(0, 10 ** 30)
        vals.append((D(mag) * (1 if rng.random() < 0.5 else -1))
                    .scaleb(-10))
    df = daft.from_pydict({"k": keys,
                           "v": _mk(vals).to_pylist()})  # keep Decimals
    out = df.groupby("k").agg(
        col("v").sum().alias("s"),
        col("v").min().alias("mn"),
        col("v").max().alias("mx")).sort("k").to_pydict()
    import collections
    want = collections.defaultdict(list)
    for k, v in zip(keys, vals):
        want[k].append(v)
    for i, k in enumerate(out["k"]):
        assert out["s"][i] == sum(want[k]), f"group {k} sum"
        assert out["mn"][i] == min(want[k])
        assert out["mx"][i] == max(want[k])


def test_global_sum():
    vals = _rand_wide(4096, digits=30, seed=7)
    df = daft.from_pydict({"v": vals})
    got = df.agg(col("v").sum().alias("s")).to_pydict()["s"][0]
    assert got == sum(vals)


def test_casts():
    w = _mk([D("123456789012345678901.123456789"), None],
            DataType.decimal128(30, 9))
    # rescale wider
    up = w.cast(DataType.decimal128(32, 11))
    assert up.to_pylist()[0] == D("123456789012345678901.12345678900")
    # downscale with rounding
    dn = w.cast(DataType.decimal128(25, 2))
    assert dn.to_pylist()[0] == D("123456789012345678901.12")
    # to float (approximate)
    f = w.cast(DataType.float64())
    assert abs(f.to_pylist()[0] - 1.2345678901234568e20) < 1e6
    # narrow fit
    small = _mk([D("12.3456789012")], WIDE)
    nar = small.cast(DataType.decimal128(18, 6))
    assert nar.to_pylist() == [D("12.345679")]
    # narrow overflow raises
    with pytest.raises(ValueError):
        _mk([D(10 ** 30).scaleb(-10)]).cast(DataType.decimal128(18, 10))
    # int -> wide
    iw = Series.from_pylist("i", [12345, -7], DataType.int64()) \
        .cast(DataType.decimal128(25, 4))
    assert iw.to_pylist() == [D("12345.0000"), D("-7.0000")]


def test_arrow_roundtrip():
    import pyarrow as pa
    vals = _rand_wide(64, seed=8) + [None]
    s = _mk(vals)
    arr = s.to_arrow()
    assert pa.types.is_decimal(arr.type) and arr.type.precision == 38
    assert arr.to_pylist() == vals
    from daft_amd.arrow_interop import from_arrow_array
    back = from_arrow_array("d", pa.array(vals, pa.decimal128(38, 10)))
    assert back.to_pylist() == vals


def test_parquet_roundtrip(tmp_path):
    vals = _rand_wide(128, seed=9)
    df = daft.from_pydict({"d": vals, "k": list(range(128))})
    p = str(tmp_path / "w")
    df.write_parquet(p)
    back = daft.read_parquet(p + "/**/*.parquet").sort("k").to_pydict()
    assert back["d"] == vals


def test_wide_group_and_join_keys_cpu():
    """CPU grouping/joining on wide keys runs the exact host path
    (hashable python Decimals); the GPU hash kernels refuse wide keys
    with a clear NotImplementedError (kernels _descs / hash_columns)."""
    a = D("12345678901234567890123456.789")
    vals = [a, D("2.5"), a, D("2.5"), D("-7.1")]
    df = daft.from_pydict({"d": vals, "x": [1, 2, 3, 4, 5]})
    out = df.groupby("d").agg(col("x").sum().alias("s")).sort("s") \
        .to_pydict()
    assert out["s"] == [4, 5, 6]
    df2 = daft.from_pydict({"d": [D("2.5"), D("-7.1")], "y": [10, 20]})
    j = df.join(df2, on="d").sort("x").to_pydict()
    assert j["x"] == [2, 4, 5] and j["y"] == [10, 10, 20]


def test_wide_hash_partition_raises():
    from daft_amd.kernels.rowops import hash_columns
    s = _mk(_rand_wide(8))
    with pytest.raises(NotImplementedError, match="p>18"):
        hash_columns([s])
