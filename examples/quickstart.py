"""daft_amd quickstart — the reference's README tour, MI355X-native.

Run anywhere (CPU falls back automatically); on an MI355X every hot
operator below executes as a hand-written HIP kernel over HBM-resident
columns.

    python examples/quickstart.py
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))))
import daft_amd as daft
from daft_amd import col
from daft_amd.functions import format as fmt, when
from daft_amd.window import Window
from daft_amd.functions import row_number

# ---- construct --------------------------------------------------------
df = daft.from_pydict({
    "city": ["SF", "NY", "SF", "LA", "NY", "SF"],
    "fare": [12.5, 31.0, 8.25, 14.0, 22.0, 19.75],
    "tip": [2.0, 5.5, 1.0, 2.5, 4.0, 3.0],
})

# ---- expressions / filters / aggregates ------------------------------
out = (df
       .with_column("total", col("fare") + col("tip"))
       .with_column("bucket", when(col("fare") > 20, "high")
                    .when(col("fare") > 10, "mid").otherwise("low"))
       .where(col("total") > 10)
       .groupby("city")
       .agg(col("total").sum().alias("revenue"),
            col("total").mean().alias("avg_total"),
            col("tip").max().alias("best_tip"))
       .sort("revenue", desc=True))
print(out.to_pydict())

# ---- SQL (same engine) ------------------------------------------------
print(daft.sql("select city, count(*) as rides from df "
               "group by city order by rides desc").to_pydict())

# ---- window functions -------------------------------------------------
w = Window().partition_by("city").order_by("fare")
print(df.with_window_columns({"rank_in_city": row_number().over(w)})
      .sort(["city", "fare"]).to_pydict())

# ---- joins ------------------------------------------------------------
zones = daft.from_pydict({"city": ["SF", "NY", "LA"],
                          "zone": ["west", "east", "west"]})
print(df.join(zones, on="city")
      .groupby("zone").agg(col("fare").sum().alias("fares"))
      .sort("zone").to_pydict())

# ---- files ------------------------------------------------------------
import tempfile, os
d = tempfile.mkdtemp()
df.write_parquet(os.path.join(d, "rides"))
back = daft.read_parquet(os.path.join(d, "rides"))
assert back.count_rows() == df.count_rows()
print("parquet round-trip:", back.count_rows(), "rows")
print("done.")

# ---- round-2 additions ------------------------------------------------
# object storage (S3-compatible; works against minio or AWS):
#   cfg = daft.IOConfig(s3=daft.io.object_store.S3Config(
#       endpoint_url="http://minio:9000", key_id=..., access_key=...))
#   daft.read_parquet("s3://bucket/tbl/**/*.parquet", io_config=cfg)
#   df.write_parquet("s3://bucket/out/")          # multipart upload
# hive-partitioned reads:
#   daft.read_parquet("data/year=*/region=*/*.parquet")
# checkpoint/resume on an object store:
#   store = daft.checkpoint.ObjectStoreCheckpointStore("s3://b/ckpt/", cfg)
# out-of-core: host-resident tables stream through HBM automatically;
# jq-style JSON filters:
#   daft.functions.jq(col("j"), ".user.tags[]")
# observability:
#   daft_amd.dashboard.serve(8238)                # live query browser
#   from daft_amd.subscribers.otlp import OTLPFileSpanExporter
