"""daft_amd — an MI355X-native distributed DataFrame / query engine.

A from-scratch framework with the capabilities of the reference DataFrame
engine (lazy DataFrame/SQL API, logical plan + optimizer, streaming
vectorized execution, multimodal columns), re-designed GPU-first:

* columnar data lives in HBM3E as torch tensors (Arrow-compatible layouts)
* hot physical operators (hash join, hash groupby-agg, radix sort,
  filter/take, partition-scatter) are hand-written HIP/CDNA4 kernels (csrc/)
* distributed execution is SPMD: one process per GPU, RCCL collectives over
  xGMI for repartition/broadcast/gather exchanges (torch.distributed)
"""
__version__ = "0.1.0"

from .schema import DataType, Field, Schema, TypeKind  # noqa: F401
from .series import Series  # noqa: F401
from .recordbatch import RecordBatch  # noqa: F401
from .expressions import Expression, col, lit, element, interval  # noqa: F401
from .dataframe import DataFrame  # noqa: F401
from .context import (  # noqa: F401
    get_context, set_runner_native, set_runner_distributed,
    set_execution_config, execution_config_ctx, set_planning_config,
    attach_subscriber, detach_subscriber,
)
from .io import (  # noqa: F401
    from_pydict, from_arrow, from_pandas, from_glob_path,
    read_parquet, read_csv, read_json, read_text, read_warc,
    read_sql, read_blob, from_glob_path, from_files, read_deltalake,
    read_iceberg, read_lance, read_hudi, read_kafka, read_mcap,
    read_paimon, read_huggingface, read_video_frames,
)
from .catalog import Catalog, Identifier, MemoryCatalog, Session, \
    current_session  # noqa: F401
from .window import Window  # noqa: F401
from .checkpoint import (CheckpointConfig, CheckpointStore,  # noqa: F401
                         LocalCheckpointStore, MemoryCheckpointStore)
from .sql import sql  # noqa: F401
from .udf import func, udf, cls, method, udaf  # noqa: F401
from .functions import coalesce  # noqa: F401


def from_pylist(rows, device=None):
    """Build a DataFrame from a list of row dicts."""
    keys = []
    for r in rows:
        for k in r:
            if k not in keys:
                keys.append(k)
    data = {k: [r.get(k) for r in rows] for k in keys}
    return from_pydict(data, device=device)

from . import datasets  # noqa: E402
from .ext import load_extension, ext_function  # noqa: E402
from .file import File  # noqa: E402,F401
