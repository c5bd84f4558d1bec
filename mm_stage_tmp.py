import io, time, numpy as np, torch, cProfile, pstats
import daft_amd as daft
from daft_amd import col
from benchmarks.bench_multimodal import synth_pngs
from daft_amd.functions.ai import embed_image
pngs = synth_pngs(256)
codes = np.random.RandomState(1).randint(0, 256, 100_000)
vals = [pngs[c] for c in codes]
df = daft.from_pydict({"data": vals}, device="cpu").into_batches(32768)
def pipe():
    return (df.with_column("img", col("data").image.decode())
            .with_column("small", col("img").image.resize(224, 224))
            .with_column("t", col("small").image.to_tensor())
            .select(embed_image(col("t"), provider="torch", dimensions=512).alias("emb"))
            .count_rows())
t0=time.time(); n = pipe(); torch.cuda.synchronize()
print(f"full: {100000/(time.time()-t0):.0f} img/s", flush=True)
pr = cProfile.Profile(); pr.enable()
pipe(); torch.cuda.synchronize()
pr.disable()
import io as _io
s = _io.StringIO(); pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(18)
print(s.getvalue()[:2600])
