import io, time, numpy as np, torch
from PIL import Image
import daft_amd as daft
from daft_amd import col
from benchmarks.bench_multimodal import synth_pngs
pngs = synth_pngs(256)
codes = np.random.RandomState(1).randint(0, 256, 100_000)
vals = [pngs[c] for c in codes]
df = daft.from_pydict({"data": vals}, device="cpu").into_batches(32768)
torch.cuda.synchronize()
def stage(label, fn):
    t0=time.time(); out = fn(); torch.cuda.synchronize()
    print(f"{label}: {100_000/(time.time()-t0):.0f} img/s ({time.time()-t0:.1f}s)", flush=True)
    return out
d1 = stage("decode", lambda: df.with_column("img", col("data").image.decode()).collect())
d2 = stage("resize", lambda: d1.with_column("small", col("img").image.resize(224,224)).collect())
d3 = stage("to_tensor", lambda: d2.select(col("small").image.to_tensor().alias("t")).collect())
from daft_amd.functions.ai import embed_image
d4 = stage("embed", lambda: d3.select(embed_image(col("t"), provider="torch", dimensions=512).alias("e")).collect())
