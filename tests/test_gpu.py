"""GPU tests: every HIP kernel vs its CPU fallback / plain-torch reference.
All run on a real MI355X via gpurun (marked gpu)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)

import daft_amd as daft
from daft_amd import DataType, Series, col, lit
from daft_amd.kernels import native_required, rowops
from daft_amd.kernels import strings as strk


@pytest.fixture(scope="module", autouse=True)
def _require_native():
    # the HIP extension must be present on a GPU box — no silent fallback
    native_required()


def _gpu(s: Series) -> Series:
    return s.to("cuda:0")


def test_native_loaded():
    mod = native_required()
    assert mod.__name__.endswith("_native")


def test_compact_indices_matches_cpu():
    torch.manual_seed(1)
    mask_cpu = torch.rand(1_000_003) < 0.3
    m_gpu = Series("m", DataType.bool(), data=mask_cpu.cuda())
    m_cpu = Series("m", DataType.bool(), data=mask_cpu)
    from daft_amd import kernels
    got = kernels.compact_indices(m_gpu).cpu()
    want = kernels.compact_indices(m_cpu)
    assert torch.equal(got, want)


def test_hash_rows_groups_match_cpu():
    torch.manual_seed(2)
    vals = torch.randint(0, 1000, (100_000,))
    s_gpu = Series("a", DataType.int64(), data=vals.cuda())
    h = rowops.hash_columns([s_gpu]).cpu()
    # same value -> same hash
    idx = torch.argsort(vals, stable=True)
    hv = h[idx]
    vv = vals[idx]
    same_val = vv[1:] == vv[:-1]
    same_hash = hv[1:] == hv[:-1]
    assert torch.equal(same_val, same_val & same_hash)
    # distinct hash count is plausible (no mass collisions)
    assert len(torch.unique(h)) == len(torch.unique(vals))


def test_groupby_gpu_matches_cpu():
    torch.manual_seed(3)
    n = 500_000
    k1 = torch.randint(0, 997, (n,))
    k2 = torch.randint(0, 5, (n,))
    v = torch.rand(n, dtype=torch.float64)
    df_gpu = daft.from_pydict({"k1": k1, "k2": k2, "v": v}, device="cuda:0")
    df_cpu = daft.from_pydict({"k1": k1, "k2": k2, "v": v}, device="cpu")
    q = lambda df: df.groupby("k1", "k2").agg(
        col("v").sum().alias("s"), col("v").count().alias("c"),
        col("v").min().alias("mn"), col("v").max().alias("mx"),
        col("v").mean().alias("avg"),
    ).sort(["k1", "k2"]).to_pydict()
    got, want = q(df_gpu), q(df_cpu)
    assert got["k1"] == want["k1"] and got["k2"] == want["k2"]
    assert got["c"] == want["c"]
    for key in ("s", "mn", "mx", "avg"):
        assert got[key] == pytest.approx(want[key], rel=1e-9)


def test_groupby_string_keys():
    n = 200_000
    torch.manual_seed(4)
    ks = [f"key_{int(i)}" for i in torch.randint(0, 50, (n,))]
    v = torch.arange(n, dtype=torch.int64)
    g = daft.from_pydict({"k": ks, "v": v}, device="cuda:0") \
        .groupby("k").agg(col("v").sum().alias("s")).sort("k").to_pydict()
    c = daft.from_pydict({"k": ks, "v": v}, device="cpu") \
        .groupby("k").agg(col("v").sum().alias("s")).sort("k").to_pydict()
    assert g == c


def test_join_gpu_matches_cpu():
    torch.manual_seed(5)
    nl, nr = 300_000, 50_000
    lk = torch.randint(0, 60_000, (nl,))
    rk = torch.randperm(60_000)[:nr]
    rv = torch.rand(nr, dtype=torch.float64)
    for how in ("inner", "left", "semi", "anti"):
        lg = daft.from_pydict({"k": lk}, device="cuda:0")
        rg = daft.from_pydict({"k": rk, "v": rv}, device="cuda:0")
        lc = daft.from_pydict({"k": lk}, device="cpu")
        rc = daft.from_pydict({"k": rk, "v": rv}, device="cpu")
        got = lg.join(rg, on="k", how=how).sort(
            ["k"] + (["v"] if how in ("inner", "left") else [])).to_pydict()
        want = lc.join(rc, on="k", how=how).sort(
            ["k"] + (["v"] if how in ("inner", "left") else [])).to_pydict()
        assert got == want, how


def test_join_string_keys_gpu():
    lk = ["a", "bb", "ccc", "bb", "zz"]
    rk = ["bb", "ccc", "q"]
    rv = [1, 2, 3]
    g = daft.from_pydict({"k": lk}, device="cuda:0").join(
        daft.from_pydict({"k": rk, "v": rv}, device="cuda:0"), on="k") \
        .sort(["k"]).to_pydict()
    assert g == {"k": ["bb", "bb", "ccc"], "v": [1, 1, 2]}


def test_radix_argsort_vs_torch():
    torch.manual_seed(6)
    for n in (1, 63, 64, 1000, 1_000_000):
        keys = torch.randint(-(2**62), 2**62, (n,)).cuda()
        s = Series("k", DataType.int64(), data=keys)
        perm = rowops.argsort_multi([s], [False], [False]).cpu()
        want = torch.argsort(keys.cpu(), stable=True)
        assert torch.equal(keys.cpu()[perm], keys.cpu()[want])


def test_radix_argsort_stability():
    # equal keys must keep input order
    keys = torch.tensor([5, 1, 5, 1, 5], dtype=torch.int64).cuda()
    perm = native_required().radix_argsort(
        keys)  # raw kernel: unsigned order
    assert perm.cpu().tolist() == [1, 3, 0, 2, 4]


def test_sort_floats_and_nulls_gpu():
    vals = [3.5, None, -1.0, float("nan"), 0.0, None, -0.0, 100.25]
    g = daft.from_pydict({"v": vals}, device="cuda:0").sort("v").to_pydict()
    c = daft.from_pydict({"v": vals}, device="cpu").sort("v").to_pydict()
    assert str(g) == str(c)  # str() so NaN placement compares equal


def test_sort_strings_gpu():
    import random
    random.seed(7)
    alphabet = "abcdefg"
    vals = ["".join(random.choice(alphabet)
                    for _ in range(random.randint(0, 20)))
            for _ in range(20_000)]
    g = daft.from_pydict({"s": vals}, device="cuda:0").sort("s").to_pydict()
    assert g["s"] == sorted(vals)
    g2 = daft.from_pydict({"s": vals}, device="cuda:0") \
        .sort("s", desc=True).to_pydict()
    assert g2["s"] == sorted(vals, reverse=True)


def test_multi_key_sort_gpu():
    torch.manual_seed(8)
    a = torch.randint(0, 10, (50_000,))
    b = torch.rand(50_000, dtype=torch.float64)
    g = daft.from_pydict({"a": a, "b": b}, device="cuda:0") \
        .sort(["a", "b"], desc=[False, True]).to_pydict()
    c = daft.from_pydict({"a": a, "b": b}, device="cpu") \
        .sort(["a", "b"], desc=[False, True]).to_pydict()
    assert g == c


def test_take_strings_gpu():
    vals = ["aaa", "b", "", "dddd", None]
    s = _gpu(Series.from_pylist("s", vals))
    idx = torch.tensor([4, 3, 0, -1, 2], dtype=torch.int64).cuda()
    assert s.take(idx).to_pylist() == [None, "dddd", "aaa", None, ""]


def test_string_predicates_gpu():
    vals = ["hello world", "worldly", None, "says hello", "WORLD", ""]
    s = _gpu(Series.from_pylist("s", vals))
    assert strk.contains(s, "world").to_pylist() == \
        [True, True, None, False, False, False]
    assert strk.startswith(s, "world").to_pylist() == \
        [False, True, None, False, False, False]
    assert strk.endswith(s, "hello").to_pylist() == \
        [False, False, None, True, False, False]


def test_like_gpu():
    vals = ["PROMO BURNISHED", "SMALL PROMO", "special requests package",
            None, "xspecialyrequestsz"]
    s = _gpu(Series.from_pylist("s", vals))
    assert strk.like(s, "PROMO%").to_pylist() == \
        [True, False, False, None, False]
    assert strk.like(s, "%special%requests%").to_pylist() == \
        [False, False, True, None, True]
    assert strk.like(s, "%PROMO").to_pylist() == \
        [False, True, False, None, False]


def test_str_ops_gpu():
    vals = ["Hello", "WORLD", None]
    s = _gpu(Series.from_pylist("s", vals))
    assert strk.lower(s).to_pylist() == ["hello", "world", None]
    assert strk.upper(s).to_pylist() == ["HELLO", "WORLD", None]
    assert strk.substr(s, 1, 3).to_pylist() == ["ell", "ORL", None]
    assert strk.length(s).to_pylist() == [5, 5, None]
    assert strk.concat_str([s, s]).to_pylist() == \
        ["HelloHello", "WORLDWORLD", None]


def test_string_compare_gpu():
    a = _gpu(Series.from_pylist("a", ["a", "bb", "c"]))
    b = _gpu(Series.from_pylist("b", ["a", "ba", "d"]))
    assert a.compare(b, "eq").to_pylist() == [True, False, False]
    assert a.compare(b, "lt").to_pylist() == [False, False, True]
    assert a.compare(b, "ge").to_pylist() == [True, True, False]


def test_distinct_gpu():
    torch.manual_seed(9)
    v = torch.randint(0, 1000, (100_000,))
    g = daft.from_pydict({"v": v}, device="cuda:0").distinct().to_pydict()
    assert sorted(g["v"]) == sorted(set(v.tolist()))


def test_count_distinct_gpu():
    torch.manual_seed(10)
    gcol = torch.randint(0, 20, (100_000,))
    v = torch.randint(0, 500, (100_000,))
    got = daft.from_pydict({"g": gcol, "v": v}, device="cuda:0") \
        .groupby("g").agg(col("v").count_distinct().alias("nd")) \
        .sort("g").to_pydict()
    want = daft.from_pydict({"g": gcol, "v": v}, device="cpu") \
        .groupby("g").agg(col("v").count_distinct().alias("nd")) \
        .sort("g").to_pydict()
    assert got == want


def test_window_gpu():
    torch.manual_seed(11)
    g = torch.randint(0, 7, (10_000,))
    v = torch.rand(10_000, dtype=torch.float64)
    from daft_amd.window import Window
    from daft_amd.functions import row_number
    w = Window().partition_by("g").order_by("v")
    q = lambda df: df.with_window_columns({
        "rn": row_number().over(w),
        "sv": col("v").sum().over(w),
    }).sort(["g", "v"]).to_pydict()
    got = q(daft.from_pydict({"g": g, "v": v}, device="cuda:0"))
    want = q(daft.from_pydict({"g": g, "v": v}, device="cpu"))
    assert got["rn"] == want["rn"]
    assert got["sv"] == pytest.approx(want["sv"], rel=1e-9)


def test_pipeline_q1_shape():
    """Mini TPC-H Q1-shaped query, GPU vs CPU."""
    torch.manual_seed(12)
    n = 200_000
    data = {
        "rf": torch.randint(0, 3, (n,)),
        "ls": torch.randint(0, 2, (n,)),
        "qty": torch.randint(1, 51, (n,)).to(torch.float64),
        "price": torch.rand(n, dtype=torch.float64) * 1000,
        "disc": torch.rand(n, dtype=torch.float64) * 0.1,
        "tax": torch.rand(n, dtype=torch.float64) * 0.08,
        "shipdate": torch.randint(0, 2500, (n,)),
    }
    def q(df):
        return (df.where(col("shipdate") <= 2200)
                .groupby("rf", "ls")
                .agg(col("qty").sum().alias("sum_qty"),
                     (col("price") * (1 - col("disc"))).sum().alias("disc_price"),
                     (col("price") * (1 - col("disc")) * (1 + col("tax"))).sum().alias("charge"),
                     col("qty").mean().alias("avg_qty"),
                     col("qty").count().alias("cnt"))
                .sort(["rf", "ls"]).to_pydict())
    got = q(daft.from_pydict(data, device="cuda:0"))
    want = q(daft.from_pydict(data, device="cpu"))
    assert got["cnt"] == want["cnt"]
    for k in ("sum_qty", "disc_price", "charge", "avg_qty"):
        assert got[k] == pytest.approx(want[k], rel=1e-9)


def test_minhash_gpu_matches_cpu():
    docs = ["the quick brown fox jumps over the lazy dog",
            "pack my box with five dozen liquor jugs",
            "a b", "single", ""] * 200
    g = daft.from_pydict({"t": docs}, device="cuda:0") \
        .select(col("t").minhash(64, ngram_size=2).alias("mh")).to_pydict()
    c = daft.from_pydict({"t": docs}, device="cpu") \
        .select(col("t").minhash(64, ngram_size=2).alias("mh")).to_pydict()
    assert g == c


def test_simhash_gpu_matches_cpu():
    docs = ["the quick brown fox jumps over the lazy dog",
            "the quick brown fox jumps over the lazy cat",
            "x" * 200, "ab", "", None] * 100
    g = daft.from_pydict({"t": docs}, device="cuda:0") \
        .select(col("t").simhash(4).alias("h")).to_pydict()
    c = daft.from_pydict({"t": docs}, device="cpu") \
        .select(col("t").simhash(4).alias("h")).to_pydict()
    assert g == c


def test_hll_approx_count_distinct_gpu():
    torch.manual_seed(20)
    true_n = 5000
    v = torch.randint(0, true_n, (500_000,))
    est = daft.from_pydict({"v": v}, device="cuda:0") \
        .agg(col("v").approx_count_distinct().alias("n")).to_pydict()["n"][0]
    true_d = len(torch.unique(v))
    assert abs(est - true_d) / true_d < 0.05, (est, true_d)


def test_image_resize_gpu_close_to_cpu():
    import numpy as np
    import io
    from PIL import Image as PILImage
    rng = np.random.RandomState(0)
    imgs = []
    for _ in range(8):
        h, w = rng.randint(8, 64), rng.randint(8, 64)
        arr = rng.randint(0, 256, (h, w, 3), dtype="uint8")
        buf = io.BytesIO()
        PILImage.fromarray(arr, "RGB").save(buf, format="PNG")
        imgs.append(buf.getvalue())
    q = lambda dev: daft.from_pydict({"b": imgs}, device=dev) \
        .select(col("b").image.decode().image.resize(16, 16).alias("r")) \
        .to_pydict()["r"]
    g, c = q("cuda:0"), q("cpu")
    diffs = [abs(int(a) - int(b)) for ga, ca in zip(g, c)
             for a, b in zip(ga, ca)]
    assert sum(d <= 1 for d in diffs) / len(diffs) > 0.99, max(diffs)


def test_embed_image_gpu():
    from daft_amd.functions.ai import embed_image
    import numpy as np
    import io
    from PIL import Image as PILImage
    arr = np.full((32, 32, 3), 128, dtype="uint8")
    buf = io.BytesIO()
    PILImage.fromarray(arr, "RGB").save(buf, format="PNG")
    df = daft.from_pydict({"b": [buf.getvalue()] * 4}, device="cuda:0")
    out = (df.with_column("t", col("b").image.decode()
                          .image.resize(32, 32).image.to_tensor())
           .select(embed_image(col("t"), provider="torch",
                               dimensions=64).alias("e"))
           .to_pydict()["e"])
    assert len(out[0]) == 64


def test_dense_range_groupby_matches_hash():
    torch.manual_seed(33)
    n = 400_000
    # dense int keys (dense-range fast path) vs string keys (hash path)
    k = torch.randint(0, 50_000, (n,))
    v = torch.rand(n, dtype=torch.float64)
    got = daft.from_pydict({"k": k, "v": v}, device="cuda:0") \
        .groupby("k").agg(col("v").sum().alias("s"),
                          col("v").count().alias("c")) \
        .sort("k").to_pydict()
    want = daft.from_pydict({"k": k, "v": v}, device="cpu") \
        .groupby("k").agg(col("v").sum().alias("s"),
                          col("v").count().alias("c")) \
        .sort("k").to_pydict()
    assert got["k"] == want["k"] and got["c"] == want["c"]
    assert got["s"] == pytest.approx(want["s"], rel=1e-9)


def test_groupby_wide_range_falls_back_to_hash():
    # range too large for the dense path: must still be correct
    torch.manual_seed(34)
    k = torch.randint(-2**60, 2**60, (50_000,))
    k = torch.cat([k, k])  # ensure duplicates
    got = daft.from_pydict({"k": k}, device="cuda:0") \
        .groupby("k").agg(col("k").count().alias("c")).to_pydict()
    assert sorted(got["c"]) == sorted(
        daft.from_pydict({"k": k}, device="cpu")
        .groupby("k").agg(col("k").count().alias("c")).to_pydict()["c"])


def test_parquet_roundtrip_gpu(tmp_path):
    df = daft.from_pydict({"a": [1, 2, 3], "s": ["x", None, "zz"],
                           "f": [1.5, 2.5, None]}, device="cuda:0")
    df.write_parquet(str(tmp_path / "o"))
    back = daft.read_parquet(str(tmp_path / "o") + "/*.parquet") \
        .sort("a").to_pydict()
    assert back == {"a": [1, 2, 3], "s": ["x", None, "zz"],
                    "f": [1.5, 2.5, None]}


def test_cache_spill_and_reload_gpu():
    from daft_amd.context import get_context
    ctx = get_context()
    df = daft.from_pydict({"a": list(range(100_000))},
                          device="cuda:0").collect()
    freed = ctx.cache.spill_lru("cuda:0")
    assert freed > 0
    # spilled partitions transparently reload on the next query
    out = df.where(col("a") < 3).to_pydict()
    assert out == {"a": [0, 1, 2]}


def test_sql_on_gpu():
    from daft_amd.sql import sql
    t = daft.from_pydict({"g": ["a", "b", "a"], "v": [1.0, 2.0, 3.0]},
                         device="cuda:0")
    out = sql("select g, sum(v) as s from t group by g order by g")
    assert out.to_pydict() == {"g": ["a", "b"], "s": [4.0, 2.0]}


def test_out_of_core_host_streaming():
    """Host-resident source + tiny stream_morsel_rows: the source slices
    morsels through HBM, aggregates fold partials, joins probe per-morsel;
    results must equal the device-resident plan."""
    from daft_amd.context import get_context
    cfg = get_context().execution_config
    old = cfg.stream_morsel_rows
    try:
        cfg.stream_morsel_rows = 10_000
        n = 100_000
        data = {
            "g": [f"k{i % 7}" for i in range(n)],
            "v": [float(i % 1000) for i in range(n)],
            "k": [i % 53 for i in range(n)],
        }
        host = daft.from_pydict(data, device="cpu")
        dev = daft.from_pydict(data, device="cuda:0")
        dim = daft.from_pydict({"k": list(range(53)),
                                "w": [i * 2.0 for i in range(53)]},
                               device="cuda:0")
        q = lambda df: (df.join(dim, on="k")
                        .groupby("g")
                        .agg((col("v") * col("w")).sum().alias("s"),
                             col("v").count().alias("c"))
                        .sort("g").to_pydict())
        out_host = q(host)
        out_dev = q(dev)
        assert out_host["g"] == out_dev["g"]
        assert out_host["c"] == out_dev["c"]
        assert out_host["s"] == pytest.approx(out_dev["s"], rel=1e-12)
    finally:
        cfg.stream_morsel_rows = old


def test_decimal_exact_gpu():
    from decimal import Decimal as D
    vals = [D("0.10")] * 1000 + [D("0.05"), None]
    g = daft.from_pydict({"v": vals}, device="cuda:0")
    tot = g.agg(col("v").sum().alias("s")).to_pydict()["s"][0]
    assert tot == D("100.05")
    # groupby + join + sort on decimal keys (int64 W8 row-op path)
    l = daft.from_pydict({"k": [D("2.50"), D("1.10"), D("2.50")] * 100},
                         device="cuda:0")
    r = daft.from_pydict({"k": [D("1.10"), D("2.50")], "w": [1, 2]},
                         device="cuda:0")
    j = l.join(r, on="k").groupby("k").agg(col("w").sum().alias("t")) \
        .sort("k").to_pydict()
    assert j["t"] == [100, 400]
    srt = daft.from_pydict({"v": [D("3.3"), D("1.1"), D("2.2")]},
                           device="cuda:0").sort("v").to_pydict()["v"]
    assert srt == [D("1.1"), D("2.2"), D("3.3")]


def test_fused_filter_aggregate_gpu_matches_cpu():
    """High-selectivity filters fuse into the aggregation as a row mask
    on GPU; results must match the CPU (compacted) plan."""
    torch.manual_seed(33)
    n = 300_000
    data = {
        "g": [f"k{i % 9}" for i in range(n)],
        "v": torch.rand(n, dtype=torch.float64),
        "w": torch.randint(0, 100, (n,)),
    }
    q = lambda df: (df.where(col("w") >= 5)     # ~95% selectivity: fused
                    .groupby("g")
                    .agg(col("v").sum().alias("s"),
                         col("v").count().alias("c"),
                         col("v").min().alias("mn"),
                         col("v").max().alias("mx"),
                         col("w").mean().alias("mu"))
                    .sort("g").to_pydict())
    got = q(daft.from_pydict(data, device="cuda:0"))
    want = q(daft.from_pydict(data, device="cpu"))
    assert got["g"] == want["g"] and got["c"] == want["c"]
    for k in ("s", "mn", "mx", "mu"):
        assert got[k] == pytest.approx(want[k], rel=1e-12)
    # low selectivity takes the compacted path: same results
    q2 = lambda df: (df.where(col("w") < 5).groupby("g")
                     .agg(col("v").sum().alias("s")).sort("g").to_pydict())
    assert q2(daft.from_pydict(data, device="cuda:0"))["s"] == \
        pytest.approx(q2(daft.from_pydict(data, device="cpu"))["s"],
                      rel=1e-12)


def test_multi_agg_fused_kernel_matches_cpu():
    """grouped_multi_agg: one-pass fused sums/mins/maxes/counts vs the
    CPU per-agg reference."""
    torch.manual_seed(44)
    n = 400_000
    data = {
        "g": [f"k{i % 5}" for i in range(n)],
        "v": torch.rand(n, dtype=torch.float64),
        "w": torch.rand(n, dtype=torch.float64) * 100,
    }
    q = lambda df: (df.groupby("g")
                    .agg(col("v").sum().alias("s"),
                         col("w").sum().alias("sw"),
                         col("v").min().alias("mn"),
                         col("v").max().alias("mx"),
                         col("w").mean().alias("mu"),
                         col("v").count().alias("c"))
                    .sort("g").to_pydict())
    got = q(daft.from_pydict(data, device="cuda:0"))
    want = q(daft.from_pydict(data, device="cpu"))
    assert got["g"] == want["g"] and got["c"] == want["c"]
    for k in ("s", "sw", "mn", "mx", "mu"):
        assert got[k] == pytest.approx(want[k], rel=1e-12), k
    # with nulls: validity flows through the fused kernel
    vals = [None if i % 7 == 0 else float(i % 100) for i in range(5000)]
    d2 = {"g": [f"x{i % 3}" for i in range(5000)], "v": vals}
    q2 = lambda df: (df.groupby("g")
                     .agg(col("v").sum().alias("s"),
                          col("v").count().alias("c"),
                          col("v").min().alias("mn"))
                     .sort("g").to_pydict())
    assert q2(daft.from_pydict(d2, device="cuda:0")) == \
        q2(daft.from_pydict(d2, device="cpu"))


def test_hip_extension_plugin_on_device(tmp_path):
    """A hipcc-built plugin launches its own gfx950 kernel on the
    engine's HBM-resident column buffers through the C ABI."""
    import os
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(daft.__file__)))
    src = os.path.join(repo, "examples", "ext_plugin", "hip_plugin.hip")
    so = str(tmp_path / "hip_plugin.so")
    subprocess.run(
        ["hipcc", "--offload-arch=gfx950", "-O2", "-shared", "-fPIC",
         "-I" + os.path.join(repo, "daft_amd", "ext"), src, "-o", so],
        check=True, capture_output=True)
    names = daft.load_extension(so)
    assert "ext_saxpy" in names
    df = daft.from_pydict({"x": [1.0, 2.5, -3.0]}, device="cuda:0")
    out = df.select(daft.ext_function("ext_saxpy", col("x")).alias("y")) \
        .to_pydict()["y"]
    assert out == [3.0, 6.0, -5.0]


def test_sql_windows_and_setops_on_device():
    """SQL windows/set-ops execute on device-resident tables (the HIP
    sort/groupby kernels under the window machinery)."""
    a = daft.from_pydict({"g": ["a", "a", "b", "b", "b"] * 200,
                          "v": list(range(1000))}, device="cuda:0")
    o = daft.sql("select g, v, row_number() over (partition by g "
                 "order by v) as rn, sum(v) over (partition by g) as t "
                 "from a order by g, v limit 3").to_pydict()
    assert o["rn"] == [1, 2, 3]
    b = daft.from_pydict({"v": list(range(500, 1500))}, device="cuda:0")
    u = daft.sql("select v from a union select v from b").count_rows()
    assert u == 1500
    i = daft.sql("select v from a intersect select v from b").count_rows()
    assert i == 500


@pytest.mark.gpu
def test_fused_expr_eval_matches_unfused():
    """Fused interpreter kernel vs per-node torch evaluation across
    arithmetic, comparisons, Kleene logic, nulls, dates, select, is_in."""
    import datetime
    import random
    random.seed(7)
    n = 100_000
    data = {
        "a": [random.uniform(-100, 100) for _ in range(n)],
        "b": [random.uniform(0.1, 10) if i % 7 else None
              for i in range(n)],
        "i": [random.randint(-1000, 1000) for _ in range(n)],
        "j": [random.randint(0, 5) if i % 11 else None for i in range(n)],
        "d": [datetime.date(1992, 1, 1) +
              datetime.timedelta(days=random.randint(0, 2500))
              for _ in range(n)],
        "f": [bool(i % 3) for i in range(n)],
    }
    df = daft.from_pydict(data).collect()
    rb = df._result[0].to("cuda:0")
    from daft_amd.expressions.expressions import resolve_exprs
    from daft_amd.physical.cse import evaluate_with_cse
    from daft_amd.kernels import fused

    exprs = resolve_exprs([
        col("a") * (lit(1) - col("b")),
        col("a") + col("b") * 2.5 - col("i"),
        (col("a") > lit(0)) & (col("b") <= lit(5.0)),
        (col("i") % 2 == 0) if False else (col("i") >= lit(0)),
        col("b").is_null(),
        col("b").fill_null(lit(-1.0)),
        col("f").if_else(col("a"), col("b")),
        col("j").is_in([1, 3, 5]),
        col("d") >= lit(datetime.date(1994, 1, 1)),
        (col("a") > 0) | (col("b") > 1),   # Kleene OR with nulls
        col("i") * 3 + 7,
    ])
    fused_out = fused.try_fuse(exprs, rb)
    assert fused_out is not None, "fusion should engage on this list"
    plain = [e.evaluate(rb) for e in exprs]
    for k, (f_s, p_s) in enumerate(zip(fused_out, plain)):
        fd = f_s.cpu().to_pylist()
        pd_ = p_s.cpu().to_pylist()
        assert len(fd) == len(pd_), f"expr {k}"
        for i in (list(range(100)) + [n - 1]):
            x, y = fd[i], pd_[i]
            if y is None:
                assert x is None, f"expr {k} row {i}: {x} != None"
            elif isinstance(y, float):
                assert x == pytest.approx(y, rel=1e-12), \
                    f"expr {k} row {i}: {x} != {y}"
            else:
                assert x == y, f"expr {k} row {i}: {x} != {y}"


@pytest.mark.gpu
def test_fused_expr_in_query_path():
    """End-to-end: q1/q6-shaped queries run through the fused kernel and
    match CPU results."""
    from benchmarks.tpch import datagen, queries
    T_g = datagen.dataframes(0.01, device="cuda:0")
    T_c = datagen.dataframes(0.01, device="cpu")
    for qi in (1, 6, 12, 14, 19):
        got = queries.run_query(qi, T_g, sf=0.01).to_pydict()
        want = queries.run_query(qi, T_c, sf=0.01).to_pydict()
        assert list(got.keys()) == list(want.keys()), f"q{qi}"
        for k in got:
            for x, y in zip(got[k], want[k]):
                if isinstance(y, float):
                    assert x == pytest.approx(y, rel=1e-9), f"q{qi}.{k}"
                else:
                    assert x == y, f"q{qi}.{k}"


@pytest.mark.gpu
def test_out_of_core_hbm_stays_bounded():
    """Streaming a host-resident table keeps peak HBM far below the
    table's size (the memory-manager guarantee behind SF1000)."""
    from benchmarks.tpch import datagen
    from benchmarks.tpch.queries import run_query
    from daft_amd.context import get_context
    get_context().execution_config.stream_morsel_rows = 1 << 24
    T = datagen.dataframes(20.0, device="cpu")  # ~120M lineitem rows, host
    # size the HOST partitions from the cache — collect() would run the
    # identity query on the GPU and materialize the table in HBM
    from daft_amd.logical import plan as lp
    node = T["lineitem"]._builder.plan
    while not isinstance(node, lp.Source):
        node = node.children[0]
    li_bytes = sum(p.size_bytes()
                   for p in get_context().cache.get(node.cache_key))
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    out = run_query(6, T, sf=5.0).to_pydict()
    peak = torch.cuda.max_memory_allocated()
    assert out["revenue"][0] > 0
    # morsel streaming must not materialize the table in HBM
    assert peak < max(li_bytes // 2, 2 << 30), (peak, li_bytes)
    get_context().execution_config.stream_morsel_rows = 1 << 26


@pytest.mark.gpu
def test_batched_udf_subprocess_device_tensors():
    """Batched subprocess UDF with tensors STAYING on device: the Series
    crosses via CUDA(dmabuf) IPC, the worker computes on cuda:0 and the
    result tensor returns without a host round trip."""
    @daft.func(return_dtype=DataType.float64(), batched=True,
               use_process=True)
    def dev_scale(x):
        assert x.data.is_cuda, "worker must receive a device tensor"
        return x.data * 3.0

    df = daft.from_pydict({"x": [1.0, 2.0, 3.0]}, device="cuda:0")
    out = df.select(dev_scale(col("x")).alias("y")).to_pydict()
    assert out["y"] == [3.0, 6.0, 9.0]


@pytest.mark.gpu
def test_segmented_grouped_agg_gpu_matches_hash():
    """Sorted-gids segmented aggregation on device equals the hash-table
    kernel path for int and float values with nulls."""
    import torch as _t
    from daft_amd.kernels import rowops
    dev = "cuda:0"
    n, G = 2_000_000, 500_000
    g = _t.sort(_t.randint(0, G, (n,), device=dev)).values
    g[:G] = _t.arange(G, device=dev)
    g = _t.sort(g).values
    vi = _t.randint(-10**6, 10**6, (n,), device=dev)
    vf = _t.rand(n, dtype=_t.float64, device=dev)
    val = _t.rand(n, device=dev) > 0.1
    for data, dt in ((vi, DataType.int64()), (vf, DataType.float64())):
        s = Series("v", dt, data=data, validity=val)
        for op in ("sum", "min", "max"):
            seg = rowops._segmented_grouped_agg(g, G, data.to(
                _t.float64 if dt.is_floating() else _t.int64), val, op, dev)
            assert seg is not None, "segmented path must engage"
            out_s, cnt_s = seg
            orig = rowops._segmented_grouped_agg
            rowops._segmented_grouped_agg = lambda *a: None
            try:
                out_h, cnt_h = rowops.grouped_agg(g, G, s, op)
            finally:
                rowops._segmented_grouped_agg = orig
            if cnt_h is not None:
                assert _t.equal(cnt_s, cnt_h), (dt, op)
            live = cnt_s > 0       # empty groups carry sentinel values
            assert _t.allclose(out_s.to(_t.float64)[live],
                               out_h.to(_t.float64)[live]), (dt, op)


@pytest.mark.gpu
def test_bpe_encode_gpu_matches_oracle():
    """Wave-per-row BPE kernel vs the python oracle over random strings
    (synthetic merge table; no tokenizer downloads offline)."""
    import random
    from daft_amd.functions.tokenize import (BPETokenizer, bpe_encode_series,
                                             _bytes_to_unicode)
    random.seed(23)
    b2u = _bytes_to_unicode()
    vocab = {b2u[b]: b for b in range(256)}
    merges = []
    nxt = 256
    # random 2-symbol merges over a small alphabet, chained
    syms = [b2u[ord(c)] for c in "abcdefgh "]
    pool = list(syms)
    for _ in range(60):
        l, r = random.choice(pool), random.choice(pool)
        if l + r in vocab:
            continue
        vocab[l + r] = nxt
        merges.append((l, r))
        pool.append(l + r)
        nxt += 1
    tok = BPETokenizer(vocab, merges)
    texts = ["".join(random.choice("abcdefgh ") for _ in
                     range(random.randint(0, 300))) for _ in range(500)]
    texts += ["a" * 5000]           # over the LDS cap: host fallback row
    texts += [None, ""]
    s = Series.from_pylist("t", texts, DataType.string()).to("cuda:0")
    out = bpe_encode_series(s, tok).cpu().to_pylist()
    for i, t in enumerate(texts):
        if t is None:
            assert out[i] is None
            continue
        want = tok.encode_py(t.encode("utf-8"))
        assert out[i] == want, (i, t[:40], out[i][:10], want[:10])


@pytest.mark.gpu
def test_external_sort_gpu_spill_matches_inmemory():
    """Device sort above the HBM budget spills runs to host, range-
    partitions into buckets, and emits the same order as the in-memory
    sort."""
    import torch as _t
    from daft_amd.context import get_context
    n = 3_000_000
    g = _t.randint(0, 1000, (n,), device="cuda:0")
    v = _t.rand(n, dtype=_t.float64, device="cuda:0")
    df = daft.from_pydict({"g": g, "v": v}, device="cuda:0") \
        .into_batches(1 << 19)
    want = df.sort(["g", "v"], desc=[True, False]).to_pydict()
    cfg = get_context().execution_config
    old = cfg.memory_limit_bytes
    cfg.memory_limit_bytes = 8 << 20        # force the spill path
    try:
        got = df.sort(["g", "v"], desc=[True, False]).to_pydict()
    finally:
        cfg.memory_limit_bytes = old
    assert got["g"] == want["g"]
    assert got["v"] == want["v"]


@pytest.mark.gpu
def test_levenshtein_gpu_matches_oracle():
    """Thread-per-pair edit-distance kernel vs the python DP, incl.
    non-ASCII and >512-byte rows that take the host fallback."""
    import random
    from daft_amd.functions.strings_extra import _lev, _lev_series
    random.seed(31)
    alpha = "abcdefgh"
    a = ["".join(random.choice(alpha) for _ in range(random.randint(0, 40)))
         for _ in range(5000)]
    b = ["".join(random.choice(alpha) for _ in range(random.randint(0, 40)))
         for _ in range(5000)]
    a += ["naïve", "x" * 600, None]
    b += ["naive", "x" * 599 + "y", "z"]
    sa = Series.from_pylist("a", a, DataType.string()).to("cuda:0")
    sb = Series.from_pylist("b", b, DataType.string()).to("cuda:0")
    out = _lev_series(sa, sb).cpu().to_pylist()
    for i, (x, y) in enumerate(zip(a, b)):
        want = None if (x is None or y is None) else _lev(x, y)
        assert out[i] == want, (i, x, y, out[i], want)


def test_wide_decimal_limb_ops_on_device():
    """Wide Decimal128 (p>18, two-int64-limb storage) on HBM: the carry
    arithmetic, comparison, sort and exact SUM paths are plain torch ops
    and must produce the same bits on device as the CPU oracle
    (kernels/decimal128.py)."""
    import decimal
    import random
    decimal.getcontext().prec = 80
    D = decimal.Decimal
    rng = random.Random(11)
    vals = [(D(rng.randint(0, 10 ** 30)) *
             (1 if rng.random() < 0.5 else -1)).scaleb(-10)
            for _ in range(2000)]
    keys = [rng.randint(0, 5) for _ in range(2000)]
    df = daft.from_pydict({"k": keys, "v": vals}, device="cuda:0")
    out = df.groupby("k").agg(
        col("v").sum().alias("s"),
        col("v").min().alias("mn"),
        col("v").max().alias("mx")).sort("k").to_pydict()
    import collections
    want = collections.defaultdict(list)
    for k, v in zip(keys, vals):
        want[k].append(v)
    for i, k in enumerate(out["k"]):
        assert out["s"][i] == sum(want[k])
        assert out["mn"][i] == min(want[k])
        assert out["mx"][i] == max(want[k])
    # exact arithmetic + sort on device (operands sized so the product
    # stays inside p38 — wider products wrap like native i128)
    wide = DataType.decimal128(38, 10)
    ma = [(D(rng.randint(0, 10 ** 17)) *
           (1 if rng.random() < 0.5 else -1)).scaleb(-10)
          for _ in range(500)]
    mb = [(D(rng.randint(0, 10 ** 17)) *
           (1 if rng.random() < 0.5 else -1)).scaleb(-10)
          for _ in range(500)]
    sa = Series.from_pylist("a", ma, wide).to("cuda:0")
    sb = Series.from_pylist("b", mb, wide).to("cuda:0")
    from daft_amd.kernels import binary_op
    got = binary_op(sa, sb, "mul").cpu().to_pylist()
    assert got == [x * y for x, y in zip(ma, mb)]
    srt = daft.from_pydict({"v": vals}, device="cuda:0") \
        .sort("v").to_pydict()["v"]
    assert srt == sorted(vals)
