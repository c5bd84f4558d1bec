"""Multi-process SPMD tests: gloo backend, world_size=2, CPU — validates the
distributed planner + exchange paths that run over RCCL on GPU boxes
(ref test pattern: the reference's LocalSwordfishWorker fake cluster,
src/daft-distributed/src/scheduling/local_worker.rs)."""
import multiprocessing as mp
import os
import pickle
import socket

import pytest
import torch

pytestmark = pytest.mark.distributed


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _run_worker(rank, world, port, fn_name, conn):
    try:
        import torch.distributed as dist
        dist.init_process_group(
            backend="gloo", rank=rank, world_size=world,
            init_method=f"tcp://127.0.0.1:{port}")
        from daft_amd.context import get_context
        from daft_amd.distributed.runner import DistributedRunner
        ctx = get_context()
        ctx.set_runner(DistributedRunner(ctx))
        fn = globals()[fn_name]
        out = fn(rank, world)
        conn.send(("ok", out))
        dist.destroy_process_group()
    except Exception as e:
        import traceback
        conn.send(("err", f"{e}\n{traceback.format_exc()}"))


def _spawn(fn_name, world=2):
    ctx = mp.get_context("spawn")
    port = _free_port()
    procs, conns = [], []
    for r in range(world):
        parent, child = ctx.Pipe()
        p = ctx.Process(target=_run_worker,
                        args=(r, world, port, fn_name, child))
        p.start()
        procs.append(p)
        conns.append(parent)
    results = []
    for p, c in zip(procs, conns):
        status, payload = c.recv() if c.poll(180) else ("err", "timeout")
        p.join(timeout=30)
        if status == "err":
            for q in procs:
                if q.is_alive():
                    q.terminate()
            raise AssertionError(payload)
        results.append(payload)
    return results


# ---------------------------------------------------------------------------
# worker bodies (run under SPMD; each asserts and returns a summary)
# ---------------------------------------------------------------------------

def _shard_df(data: dict, rank: int, world: int):
    import daft_amd as daft
    n = len(next(iter(data.values())))
    per = (n + world - 1) // world
    lo, hi = rank * per, min(n, (rank + 1) * per)
    return daft.from_pydict({k: v[lo:hi] for k, v in data.items()})


def _core_checks(rank, world):
    import daft_amd as daft
    from daft_amd import col

    data = {
        "k": [1, 2, 3, 4, 1, 2, 3, 4, 5, 6],
        "g": ["a", "b", "a", "b", "a", "b", "a", "b", "a", "b"],
        "v": [1.0, 2.0, 3.0, 4.0, 5.0, 6.0, 7.0, 8.0, 9.0, 10.0],
    }
    df = _shard_df(data, rank, world)

    # grouped agg (partial/final two-phase)
    out = df.groupby("g").agg(
        col("v").sum().alias("s"), col("v").count().alias("c"),
        col("v").mean().alias("m"), col("v").max().alias("mx"),
    ).sort("g").to_pydict()
    assert out["g"] == ["a", "b"], out
    assert out["s"] == [25.0, 30.0]
    assert out["c"] == [5, 5]
    assert out["m"] == [5.0, 6.0]
    assert out["mx"] == [9.0, 10.0]

    # approx_percentile: DDSketch partial/final across the exchange
    ap = df.groupby("g").agg(
        col("v").approx_percentile(0.5).alias("p")).sort("g").to_pydict()
    assert abs(ap["p"][0] - 5.0) / 5.0 < 0.05, ap
    assert abs(ap["p"][1] - 6.0) / 6.0 < 0.05, ap

    # ungrouped agg
    tot = df.agg(col("v").sum().alias("s"),
                 col("v").mean().alias("m")).to_pydict()
    assert tot["s"] == [55.0] and tot["m"] == [5.5]

    # count_distinct (row-exchange fallback)
    nd = df.groupby("g").agg(col("k").count_distinct().alias("nd")) \
        .sort("g").to_pydict()
    assert nd["nd"] == [3, 3], nd

    # join across shards
    dim = _shard_df({"k": [1, 2, 3, 4, 5, 6],
                     "name": ["one", "two", "three", "four", "five", "six"]},
                    rank, world)
    j = df.join(dim, on="k").groupby("name") \
        .agg(col("v").sum().alias("s")).sort("name").to_pydict()
    assert j["name"] == ["five", "four", "one", "six", "three", "two"]
    assert j["s"] == [9.0, 12.0, 6.0, 10.0, 10.0, 8.0]

    # global sort
    s = df.sort("v", desc=True).to_pydict()
    assert s["v"] == sorted(data["v"], reverse=True)

    # topn + limit
    t = df.sort("v", desc=True).limit(3).to_pydict()
    assert t["v"] == [10.0, 9.0, 8.0]

    # distinct
    d = df.select("g").distinct().sort("g").to_pydict()
    assert d["g"] == ["a", "b"]

    # count_rows
    assert df.count_rows() == 10

    # semi/anti joins
    small = _shard_df({"k": [1, 6]}, rank, world)
    semi = df.join(small, on="k", how="semi").sort("v").to_pydict()
    assert semi["k"] == [1, 1, 6]
    anti = df.join(small, on="k", how="anti").count_rows()
    assert anti == 7

    # monotonically increasing id: unique across ranks
    ids = df.add_monotonically_increasing_id("id").to_pydict()["id"]
    assert len(set(ids)) == 10

    # window over partitions
    from daft_amd.window import Window
    from daft_amd.functions import row_number
    w = Window().partition_by("g").order_by("v")
    wout = df.with_window_columns({"rn": row_number().over(w)}) \
        .sort(["g", "v"]).to_pydict()
    assert wout["rn"] == [1, 2, 3, 4, 5] * 2
    return "ok"


def _tpch_checks(rank, world):
    from benchmarks.tpch import datagen, queries
    sf = 0.01
    T = datagen.dataframes(sf, device="cpu", rank=rank, world=world)
    results = {}
    for qi in range(1, 23):
        out = queries.run_query(qi, T, sf=sf).to_pydict()
        results[qi] = out
    return pickle.dumps(results)


def test_distributed_core_ops():
    res = _spawn("_core_checks", world=2)
    assert res == ["ok", "ok"]


def test_distributed_tpch_matches_single():
    res = _spawn("_tpch_checks", world=2)
    dist_results = pickle.loads(res[0])
    # both ranks must agree (results gathered on every rank)
    assert pickle.loads(res[1]).keys() == dist_results.keys()

    import math

    def norm_rows(d):
        rows = list(zip(*d.values()))
        key = lambda r: tuple(repr(x) for x in r
                              if not isinstance(x, float))
        return sorted(rows, key=key)

    from benchmarks.tpch import datagen, queries
    T = datagen.dataframes(0.01, device="cpu")
    for qi, got in dist_results.items():
        want = queries.run_query(qi, T, sf=0.01).to_pydict()
        assert list(got.keys()) == list(want.keys()), f"q{qi} columns"
        g_rows, w_rows = norm_rows(got), norm_rows(want)
        assert len(g_rows) == len(w_rows), f"q{qi} row count"
        for gr, wr in zip(g_rows, w_rows):
            for gx, wx in zip(gr, wr):
                if isinstance(wx, float):
                    assert math.isclose(gx, wx, rel_tol=1e-9, abs_tol=1e-6), \
                        f"q{qi}: {gx} != {wx}"
                else:
                    assert gx == wx, f"q{qi}: {gx!r} != {wx!r}"


def _tpch_subset_checks(rank, world):
    from benchmarks.tpch import datagen, queries
    sf = 0.01
    T = datagen.dataframes(sf, device="cpu", rank=rank, world=world)
    results = {}
    for qi in (1, 3, 5, 9, 13, 18, 21):
        results[qi] = queries.run_query(qi, T, sf=sf).to_pydict()
    return pickle.dumps(results)


def test_distributed_world4_matches_single():
    """world_size=4 (beyond the usual 2): partition tokens, exchanges
    and two-phase aggs must hold at higher rank counts — the round-end
    8-GPU scale run exercises this same planner."""
    outs = _spawn("_tpch_subset_checks", world=4)
    per_rank = [pickle.loads(o) for o in outs]
    import math

    def norm_rows(d):
        rows = list(zip(*d.values()))
        key = lambda r: tuple(repr(x) for x in r
                              if not isinstance(x, float))
        return sorted(rows, key=key)

    from benchmarks.tpch import datagen, queries
    T = datagen.dataframes(0.01, device="cpu")
    for qi, got in per_rank[0].items():
        want = queries.run_query(qi, T, sf=0.01).to_pydict()
        for r in per_rank[1:]:
            assert r[qi] == got, f"q{qi} differs across ranks"
        assert list(got.keys()) == list(want.keys())
        g_rows, w_rows = norm_rows(got), norm_rows(want)
        assert len(g_rows) == len(w_rows), f"q{qi} rows"
        for gr, wr in zip(g_rows, w_rows):
            for gx, wx in zip(gr, wr):
                if isinstance(wx, float):
                    assert math.isclose(gx, wx, rel_tol=1e-9,
                                        abs_tol=1e-6), f"q{qi}"
                else:
                    assert gx == wx, f"q{qi}"


def _exchange_protocol_checks(rank, world):
    """Direct exercise of the packed tensor all-to-all: mixed dtypes,
    strings, nulls, dict columns, empty partitions, skew, and the chunked
    (spill) path."""
    import daft_amd as daft
    from daft_amd.distributed import comm
    from daft_amd.recordbatch import RecordBatch
    from daft_amd.series import Series
    from daft_amd.schema import DataType

    # each rank builds w parts; part p carries rows tagged (rank, p, i)
    w = world
    parts = []
    for p in range(w):
        n = 3 + ((rank + p) % 2)          # ragged sizes
        ints = [rank * 1000 + p * 10 + i for i in range(n)]
        strs = [f"r{rank}p{p}i{i}" if i % 3 else None for i in range(n)]
        fls = [float(i) + rank for i in range(n)]
        rb = daft.from_pydict({"i": ints, "s": strs, "f": fls}).collect()._result[0]
        parts.append(rb)
    got = comm.exchange_batches(parts)
    d = got.to_pydict()
    # every row this rank received was addressed to it (p == rank)
    assert all((v % 1000) // 10 == rank for v in d["i"]), d["i"]
    # one row group from every source rank
    srcs = sorted(set(v // 1000 for v in d["i"]))
    assert srcs == list(range(w)), srcs
    # null pattern preserved
    for iv, sv in zip(d["i"], d["s"]):
        i_within = iv % 10
        if i_within % 3 == 0:
            assert sv is None
        else:
            assert sv == f"r{iv // 1000}p{rank}i{i_within}"

    # empty partitions to everyone but rank 0 (gather pattern)
    rb = daft.from_pydict({"x": list(range(rank + 1))}).collect()._result[0]
    empty = rb.slice(0, 0)
    got2 = comm.exchange_batches(
        [rb if p == 0 else empty for p in range(w)])
    if rank == 0:
        assert len(got2) == sum(r + 1 for r in range(w))
    else:
        assert len(got2) == 0

    # skew: everyone sends everything to rank w-1, chunked via tiny budget
    got3 = comm.exchange_batches(
        [rb if p == w - 1 else empty for p in range(w)], hbm_budget=64)
    if rank == w - 1:
        assert len(got3) == sum(r + 1 for r in range(w))
        assert sorted(got3.to_pydict()["x"])[:3] == [0, 0, 0]
    else:
        assert len(got3) == 0

    # allgather (replicating a2a)
    got4 = comm.allgather_batch(rb)
    assert len(got4) == sum(r + 1 for r in range(w))
    return "ok"


def test_exchange_tensor_protocol_world4():
    assert _spawn("_exchange_protocol_checks", world=4) == ["ok"] * 4


def test_exchange_tensor_protocol_world8():
    assert _spawn("_exchange_protocol_checks", world=8) == ["ok"] * 8


def _tpch_pair_checks(rank, world):
    import os
    from benchmarks.tpch import datagen, queries
    sf = 0.005
    T = datagen.dataframes(sf, device="cpu", rank=rank, world=world)
    results = {}
    qis = tuple(int(x) for x in
                os.environ.get("DAFT_TEST_QIS", "1,5,13,21").split(","))
    for qi in qis:
        results[qi] = queries.run_query(qi, T, sf=sf).to_pydict()
    return pickle.dumps(results)


def test_distributed_world8_matches_single():
    """world_size=8 — the exact rank count of the round-end scale run."""
    outs = _spawn("_tpch_pair_checks", world=8)
    per_rank = [pickle.loads(o) for o in outs]
    import math

    def norm_rows(d):
        rows = list(zip(*d.values()))
        key = lambda r: tuple(repr(x) for x in r
                              if not isinstance(x, float))
        return sorted(rows, key=key)

    from benchmarks.tpch import datagen, queries
    T = datagen.dataframes(0.005, device="cpu")
    for qi, got in per_rank[0].items():
        want = queries.run_query(qi, T, sf=0.005).to_pydict()
        for r in per_rank[1:]:
            assert r[qi] == got, f"q{qi} differs across ranks"
        g_rows, w_rows = norm_rows(got), norm_rows(want)
        assert len(g_rows) == len(w_rows), f"q{qi} rows"
        for gr, wr in zip(g_rows, w_rows):
            for gx, wx in zip(gr, wr):
                if isinstance(wx, float):
                    assert math.isclose(gx, wx, rel_tol=1e-9,
                                        abs_tol=1e-6), f"q{qi}"
                else:
                    assert gx == wx, f"q{qi}"


def _wide_decimal_exchange_checks(rank, world):
    """Wide Decimal128 (two-limb struct physical) columns must survive
    the packed all-to-all exchange: shard rows across ranks, groupby an
    int key with exact decimal SUM — the redistribute carries the limb
    children."""
    import decimal
    decimal.getcontext().prec = 60
    import daft_amd as daft
    from daft_amd import col
    D = decimal.Decimal
    n = 64
    keys = [i % 5 for i in range(n)]
    vals = [(D(10 ** 25 + i * 7) * (1 if i % 2 else -1)).scaleb(-8)
            for i in range(n)]
    df = _shard_df({"k": keys, "v": vals}, rank, world)
    out = df.groupby("k").agg(col("v").sum().alias("s")).sort("k") \
        .to_pydict()
    import collections
    want = collections.defaultdict(D)
    for k, v in zip(keys, vals):
        want[k] += v
    assert out["k"] == sorted(set(keys))
    assert out["s"] == [want[k] for k in out["k"]], "exact wide sums"
    # raw-row redistribution: distributed sort carries the limb children
    # through the packed exchange
    srt = df.sort("k").to_pydict()
    pairs = sorted(zip(keys, [str(v) for v in vals]))
    # distributed sort returns this rank's slice; all ranks' concat is
    # the full sorted order — check via count + local monotonicity
    ks = srt["k"]
    assert all(ks[i] <= ks[i + 1] for i in range(len(ks) - 1))
    assert all(isinstance(v, decimal.Decimal) for v in srt["v"])
    return len(out["k"])


@pytest.mark.distributed
def test_wide_decimal_exchange_world2():
    assert _spawn("_wide_decimal_exchange_checks", world=2) == [5, 5]


def _nested_exchange_checks(rank, world):
    """List / struct / map columns must survive the packed exchange
    (offsets + children buffers): shard rows, redistribute via a
    groupby, and sort — nested payloads ride along."""
    import daft_amd as daft
    from daft_amd import col
    n = 48
    keys = [i % 4 for i in range(n)]
    lists = [[i, i + 1] if i % 3 else [] for i in range(n)]
    structs = [{"u": i, "v": f"s{i}"} for i in range(n)]
    df = _shard_df({"k": keys, "l": lists, "s": structs}, rank, world)
    out = df.groupby("k").agg(
        col("l").list.length().sum().alias("tot")).sort("k").to_pydict()
    want = {}
    for k, lst in zip(keys, lists):
        want[k] = want.get(k, 0) + len(lst)
    assert out["k"] == sorted(want)
    assert out["tot"] == [want[k] for k in out["k"]]
    srt = df.sort("k").to_pydict()
    ks = srt["k"]
    assert all(ks[i] <= ks[i + 1] for i in range(len(ks) - 1))
    assert all(isinstance(x, dict) and "u" in x for x in srt["s"])
    assert all(isinstance(x, list) for x in srt["l"])
    return sum(out["tot"])


@pytest.mark.distributed
def test_nested_exchange_world2():
    tot = _spawn("_nested_exchange_checks", world=2)
    assert tot[0] == tot[1] > 0
