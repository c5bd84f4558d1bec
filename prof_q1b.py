import sys, collections, traceback, time
sys.path.insert(0, "/root/repo")
import torch
calls = collections.Counter()
for fname in ("scatter_add_", "scatter_reduce_", "index_put_", "put_"):
    orig = getattr(torch.Tensor, fname)
    def mk(orig, fname):
        def f(self, *a, **k):
            if self.is_cuda and self.numel() < 100000 or \
                    (a and hasattr(a[0], "numel")):
                st = traceback.extract_stack()[-2]
                calls[f"{fname} {st.filename.split('/')[-1]}:{st.lineno} dst={self.numel()}"] += 1
            return orig(self, *a, **k)
        return f
    setattr(torch.Tensor, fname, mk(orig, fname))
from benchmarks.tpch import datagen
from benchmarks.tpch.queries import run_query
T = datagen.dataframes(100.0, device="cuda:0")
torch.cuda.synchronize()
run_query(1, T, 100.0).collect()
torch.cuda.synchronize()
calls.clear()
t0 = time.time()
run_query(1, T, 100.0).collect()
torch.cuda.synchronize()
print(f"q1: {time.time()-t0:.3f}s")
for k, v in sorted(calls.items()):
    print(v, k)
