import sys, time, torch
sys.path.insert(0, "/root/repo")
from benchmarks.tpch import datagen
from benchmarks.tpch.queries import run_query
tables = datagen.dataframes(100.0, device="cuda:0")
torch.cuda.synchronize()
run_query(1, tables, 100.0).collect()
torch.cuda.synchronize()
for i in range(3):
    t0 = time.time()
    run_query(1, tables, 100.0).collect()
    torch.cuda.synchronize()
    print(f"q1 run {i}: {time.time()-t0:.3f}s")
