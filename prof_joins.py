import sys, time, torch
sys.path.insert(0, "/root/repo")
from benchmarks.tpch import datagen
from benchmarks.tpch.queries import run_query
tables = datagen.dataframes(100.0, device="cuda:0")
torch.cuda.synchronize()
# warmup
for q in (7, 9, 21):
    run_query(q, tables, 100.0).collect()
torch.cuda.synchronize()
import ctypes
roctx = ctypes.CDLL("libroctx64.so")
for q in (7, 9, 21):
    t0 = time.time()
    roctx.roctxRangePushA(f"q{q}".encode())
    run_query(q, tables, 100.0).collect()
    torch.cuda.synchronize()
    roctx.roctxRangePop()
    print(f"q{q}: {time.time()-t0:.3f}s")
