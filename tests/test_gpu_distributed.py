"""RCCL-branch rehearsal on a single MI355X: two NCCL(=RCCL) ranks pinned
to the SAME device execute the packed tensor all-to-all end-to-end — the
exact code path the 8-GPU scale run takes over xGMI (ref:
/root/reference/src/daft-shuffles/src/shuffle_cache.rs:47-90 replaced by
direct HBM exchange).

If this RCCL build refuses duplicate devices in one communicator, the test
xfails with that message rather than faking a pass.
"""
import multiprocessing as mp
import pickle
import socket

import pytest
import torch

pytestmark = pytest.mark.gpu


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _nccl_worker(rank, world, port, fn_name, conn):
    try:
        import os
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        import torch.distributed as dist
        torch.cuda.set_device(0)
        dist.init_process_group(
            backend="nccl", rank=rank, world_size=world,
            init_method=f"tcp://127.0.0.1:{port}")
        from daft_amd.context import get_context
        from daft_amd.distributed.runner import DistributedRunner
        ctx = get_context()
        ctx.set_runner(DistributedRunner(ctx))
        fn = globals()[fn_name]
        out = fn(rank, world)
        dist.barrier()
        conn.send(("ok", out))
        dist.destroy_process_group()
    except Exception as e:
        import traceback
        conn.send(("err", f"{e}\n{traceback.format_exc()}"))


def _spawn_nccl(fn_name, world=2, timeout=240):
    ctx = mp.get_context("spawn")
    port = _free_port()
    procs, conns = [], []
    for r in range(world):
        parent, child = ctx.Pipe()
        p = ctx.Process(target=_nccl_worker,
                        args=(r, world, port, fn_name, child))
        p.start()
        procs.append(p)
        conns.append(parent)
    results = []
    for p, c in zip(procs, conns):
        status, payload = c.recv() if c.poll(timeout) else ("err", "timeout")
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
        results.append((status, payload))
    errs = [pl for st, pl in results if st == "err"]
    if errs:
        low = errs[0].lower()
        if "duplicate" in low or "same device" in low or \
                "invalid usage" in low:
            pytest.xfail(f"RCCL refuses 2 ranks on one device: {errs[0][:200]}")
        raise AssertionError(errs[0])
    return [pl for _, pl in results]


def _rccl_exchange_body(rank, world):
    import daft_amd as daft
    from daft_amd.distributed import comm

    dev = "cuda:0"
    n = 5 + rank
    ints = [rank * 100 + i for i in range(n)]
    strs = [f"r{rank}i{i}" if i % 3 else None for i in range(n)]
    rb = daft.from_pydict({"i": ints, "s": strs}).collect()._result[0].to(dev)
    w = world
    # split rows round-robin by destination
    parts = []
    for p in range(w):
        idx = torch.tensor([i for i in range(n) if i % w == p],
                           dtype=torch.int64, device=dev)
        parts.append(rb.take(idx))
    got = comm.exchange_batches(parts)
    assert str(got.device).startswith("cuda")
    d = got.cpu().to_pydict()
    # every received row index ≡ my rank (mod world), one group per src
    assert all((v % 100) % w == rank for v in d["i"]), d["i"]
    assert sorted(set(v // 100 for v in d["i"])) == list(range(w))
    for iv, sv in zip(d["i"], d["s"]):
        i = iv % 100
        assert (sv is None) == (i % 3 == 0)
        if sv is not None:
            assert sv == f"r{iv // 100}i{i}"

    # chunked (spill) path: tiny budget forces host staging through
    # _chunked_exchange on the NCCL branch
    got2 = comm.exchange_batches(parts, hbm_budget=64)
    assert len(got2) == len(got)
    assert sorted(got2.cpu().to_pydict()["i"]) == sorted(d["i"])

    # allgather (replicating a2a) on device
    got3 = comm.allgather_batch(rb)
    assert len(got3) == sum(5 + r for r in range(w))
    return "ok"


def _rccl_tpch_body(rank, world):
    from benchmarks.tpch import datagen, queries
    sf = 0.01
    T = datagen.dataframes(sf, device="cuda:0", rank=rank, world=world)
    results = {}
    for qi in (1, 3, 5, 9, 13, 18, 21):
        results[qi] = queries.run_query(qi, T, sf=sf).to_pydict()
    return pickle.dumps(results)


def test_rccl_exchange_two_ranks_one_gpu():
    assert _spawn_nccl("_rccl_exchange_body", world=2) == ["ok", "ok"]


def test_rccl_tpch_two_ranks_matches_single():
    outs = _spawn_nccl("_rccl_tpch_body", world=2, timeout=420)
    per_rank = [pickle.loads(o) for o in outs]
    import math

    from benchmarks.tpch import datagen, queries
    T = datagen.dataframes(0.01, device="cuda:0")

    def norm_rows(d):
        rows = list(zip(*d.values()))
        key = lambda r: tuple(repr(x) for x in r
                              if not isinstance(x, float))
        return sorted(rows, key=key)

    for qi, got in per_rank[0].items():
        want = queries.run_query(qi, T, sf=0.01).to_pydict()
        assert per_rank[1][qi] == got, f"q{qi} differs across ranks"
        g_rows, w_rows = norm_rows(got), norm_rows(want)
        assert len(g_rows) == len(w_rows), f"q{qi} rows"
        for gr, wr in zip(g_rows, w_rows):
            for gx, wx in zip(gr, wr):
                if isinstance(wx, float):
                    assert math.isclose(gx, wx, rel_tol=1e-6,
                                        abs_tol=1e-5), f"q{qi}: {gx} != {wx}"
                else:
                    assert gx == wx, f"q{qi}"
