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
    read_parquet, read_csv, read_json, read_ipc, read_text, read_warc,
    read_sql, read_blob, from_glob_path, from_files, read_deltalake,
    read_iceberg, read_lance, read_hudi, read_kafka, read_mcap,
    read_paimon, read_huggingface, read_video_frames,
)
from .catalog import Catalog, Identifier, MemoryCatalog, Session, \
    current_session  # noqa: F401
from .window import Window  # noqa: F401
from .checkpoint import (CheckpointConfig, CheckpointStore,  # noqa: F401
                         LocalCheckpointStore, MemoryCheckpointStore)
from .sql import sql, sql_expr  # noqa: F401
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

from .session_api import (  # noqa: E402,F401
    TimeUnit, ImageMode, ImageFormat, ImageProperty, MediaType,
    UnionMode, IOConfig, ResourceRequest, KeyFilteringSettings,
    IdempotentCommit, Table, AudioFile, VideoFile, Hdf5File, ImageFile,
    session, set_session, attach, attach_catalog, detach_catalog,
    set_catalog, current_catalog, list_catalogs, has_catalog,
    get_catalog, create_temp_table, create_temp_view, attach_table,
    attach_view, create_table, create_table_if_not_exists, drop_table,
    detach_table, get_table, read_table, write_table, has_table,
    list_tables, create_namespace, create_namespace_if_not_exists,
    drop_namespace, has_namespace, set_namespace, current_namespace,
    attach_function, detach_function, get_function,
    get_aggregate_function, attach_provider, detach_provider,
    get_provider, has_provider, current_provider, set_provider,
    set_model, current_model, concat, open_file, metrics,
    get_loaded_extension_paths, with_subscriber, register_viz_hook,
    refresh_logger, planning_config_ctx, runners, get_or_create_runner,
    get_or_infer_runner_type, set_runner_ray, from_dask_dataframe,
    from_ray_dataset, list_namespaces)
from .session_api import range  # noqa: E402,F401,A004
