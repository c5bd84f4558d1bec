"""Top-level session/catalog convenience API (capability of the
reference's daft/__init__.py exports around Session/Catalog —
attach/create/list/drop helpers that operate on the ambient session —
plus small config/value types: TimeUnit, ImageMode, IOConfig, range)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from .catalog import Catalog, Session, current_session


# -- small value/config types ------------------------------------------------

class TimeUnit:
    """Arrow-style time unit (ref: daft.TimeUnit)."""

    def __init__(self, unit: str):
        if unit not in ("s", "ms", "us", "ns"):
            raise ValueError(f"bad time unit {unit!r}")
        self._unit = unit

    @staticmethod
    def s(): return TimeUnit("s")
    @staticmethod
    def ms(): return TimeUnit("ms")
    @staticmethod
    def us(): return TimeUnit("us")
    @staticmethod
    def ns(): return TimeUnit("ns")

    def __str__(self): return self._unit
    def __repr__(self): return f"TimeUnit({self._unit})"
    def __eq__(self, o): return str(self) == str(o)
    def __hash__(self): return hash(self._unit)


class ImageMode:
    L = "L"
    LA = "LA"
    RGB = "RGB"
    RGBA = "RGBA"


class ImageFormat:
    PNG = "PNG"
    JPEG = "JPEG"
    BMP = "BMP"
    GIF = "GIF"
    TIFF = "TIFF"


class ImageProperty:
    HEIGHT = "height"
    WIDTH = "width"
    CHANNEL = "channel"
    MODE = "mode"


class MediaType:
    IMAGE = "image"
    AUDIO = "audio"
    VIDEO = "video"


class UnionMode:
    SPARSE = "sparse"
    DENSE = "dense"


# real object-store config (S3-compatible + HTTP backends implemented;
# GCS/Azure accepted for API compatibility, raise on use)
from .io.object_store import HTTPConfig, IOConfig, S3Config  # noqa: F401


@dataclass
class ResourceRequest:
    num_cpus: Optional[float] = None
    num_gpus: Optional[float] = None
    memory_bytes: Optional[int] = None


@dataclass
class KeyFilteringSettings:
    enabled: bool = True
    threshold: float = 0.1


@dataclass
class IdempotentCommit:
    key: str = ""
    store: Any = None


class Table:
    """Catalog table handle: read() -> DataFrame, append/overwrite via
    the owning catalog (ref: daft.Table)."""

    def __init__(self, name: str, df):
        self.name = name
        self._df = df

    def read(self):
        return self._df

    def to_df(self):
        return self._df

    def __repr__(self):
        return f"Table({self.name})"


# -- ambient-session helpers --------------------------------------------------

def session() -> Session:
    return current_session()


def set_session(s: Session) -> None:
    from . import catalog as _c
    _c._session = s


def attach_catalog(catalog, alias: Optional[str] = None):
    current_session().attach_catalog(catalog, alias)
    return catalog


attach = attach_catalog


def detach_catalog(name: str) -> None:
    current_session().detach_catalog(name)


def set_catalog(name: str) -> None:
    current_session().set_catalog(name)


def current_catalog():
    return current_session().current_catalog()


def list_catalogs() -> List[str]:
    return list(current_session()._catalogs)


def has_catalog(name: str) -> bool:
    return name in current_session()._catalogs


def get_catalog(name: str):
    return current_session()._catalogs[name]


def create_temp_table(name: str, df):
    current_session().create_temp_table(name, df)
    return Table(name, df)


create_temp_view = create_temp_table
attach_table = create_temp_table
attach_view = create_temp_table


def create_table(name: str, df):
    s = current_session()
    cat = s.current_catalog()
    if cat is None:
        s.create_temp_table(name, df)
    else:
        cat.create_table(name, df)
    return Table(name, df)


def create_table_if_not_exists(name: str, df):
    try:
        if has_table(name):
            return Table(name, get_table(name))
    except Exception:
        pass
    return create_table(name, df)


def drop_table(name: str) -> None:
    s = current_session()
    try:
        s._temp.drop_table(name)
        return
    except Exception:
        pass
    cat = s.current_catalog()
    if cat is not None:
        cat.drop_table(name)


def detach_table(name: str) -> None:
    drop_table(name)


def get_table(name: str):
    return current_session().get_table(name)


def read_table(name: str):
    return get_table(name)


def write_table(name: str, df, mode: str = "append"):
    s = current_session()
    try:
        existing = s.get_table(name)
    except Exception:
        existing = None
    if existing is None or mode == "overwrite":
        s.create_temp_table(name, df.collect())
    else:
        s.create_temp_table(name, existing.concat(df).collect())


def has_table(name: str) -> bool:
    try:
        current_session().get_table(name)
        return True
    except Exception:
        return False


def list_tables(pattern: Optional[str] = None) -> List[str]:
    out = current_session().list_tables()
    if pattern:
        out = [t for t in out if pattern in t]
    return out


def list_namespaces(pattern: Optional[str] = None) -> List[str]:
    """Available namespaces in the current session (ref:
    daft/session.py list_namespaces)."""
    out = sorted(current_session().options.get("namespaces", set()))
    if pattern:
        out = [n for n in out if pattern in n]
    return out


# namespaces (flat in the memory catalog: namespace == name prefix)
def create_namespace(name: str) -> None:
    current_session().options.setdefault("namespaces", set()).add(name)


def create_namespace_if_not_exists(name: str) -> None:
    create_namespace(name)


def drop_namespace(name: str) -> None:
    current_session().options.get("namespaces", set()).discard(name)


def has_namespace(name: str) -> bool:
    return name in current_session().options.get("namespaces", set())


def set_namespace(name: str) -> None:
    current_session().options["namespace"] = name


def current_namespace() -> Optional[str]:
    return current_session().options.get("namespace")


# -- functions / providers / models -------------------------------------------

def attach_function(fn, alias: Optional[str] = None) -> None:
    current_session().options.setdefault("functions", {})[
        alias or getattr(fn, "__name__", "fn")] = fn


def detach_function(name: str) -> None:
    current_session().options.get("functions", {}).pop(name, None)


def get_function(name: str):
    return current_session().options.get("functions", {})[name]


def get_aggregate_function(name: str):
    return get_function(name)


def attach_provider(provider, alias: Optional[str] = None) -> None:
    from .ai import register_provider
    name = alias or getattr(provider, "name", "provider")
    register_provider(name, provider)
    current_session().options.setdefault("providers", {})[name] = provider


def detach_provider(name: str) -> None:
    current_session().options.get("providers", {}).pop(name, None)


def get_provider(name: str):
    return current_session().options.get("providers", {})[name]


def has_provider(name: str) -> bool:
    return name in current_session().options.get("providers", {})


def current_provider():
    return current_session().options.get("provider")


def set_provider(name: str) -> None:
    current_session().options["provider"] = name


def set_model(name: str) -> None:
    current_session().options["model"] = name


def current_model():
    return current_session().options.get("model")


# -- misc ----------------------------------------------------------------------

def range(end: int, start: int = 0, step: int = 1,  # noqa: A001
          partitions: Optional[int] = None):
    """daft.range: an integer-range DataFrame with column `id`."""
    import builtins
    from . import from_pydict
    return from_pydict({"id": list(builtins.range(start, end, step))})


def concat(dfs: List) -> Any:
    acc = dfs[0]
    for d in dfs[1:]:
        acc = acc.concat(d)
    return acc


def open_file(path_or_bytes):
    from .file import File
    return File(path_or_bytes).open()


def metrics() -> Dict[str, Any]:
    from .context import get_context
    ctx = get_context()
    return dict(getattr(ctx, "last_stats", {}) or {})


def get_loaded_extension_paths() -> List[str]:
    from .ext import _REGISTRY
    return [getattr(lib, "_name", "<so>") for lib in _REGISTRY._libs]


def with_subscriber(sub):
    from .context import get_context
    get_context().subscribers.append(sub)
    return sub


def register_viz_hook(hook) -> None:
    current_session().options["viz_hook"] = hook


def refresh_logger() -> None:
    import logging
    logging.basicConfig()


class _PlanningCtx:
    def __init__(self, **kw):
        self.kw = kw

    def __enter__(self):
        from .context import get_context
        cfg = get_context().execution_config
        self._old = {k: getattr(cfg, k) for k in self.kw
                     if hasattr(cfg, k)}
        for k, v in self.kw.items():
            if hasattr(cfg, k):
                setattr(cfg, k, v)
        return cfg

    def __exit__(self, *a):
        from .context import get_context
        cfg = get_context().execution_config
        for k, v in self._old.items():
            setattr(cfg, k, v)


def planning_config_ctx(**kwargs):
    return _PlanningCtx(**kwargs)


def runners() -> List[str]:
    return ["native", "distributed"]


def get_or_create_runner():
    from .context import get_context
    return get_context().runner()


def get_or_infer_runner_type() -> str:
    from .distributed import comm
    return "distributed" if comm.is_dist() else "native"


def set_runner_ray(*a, **k):
    raise RuntimeError("this engine scales with SPMD torch.distributed "
                       "over RCCL, not Ray; run under torchrun")


def from_dask_dataframe(*a, **k):
    raise RuntimeError("from_dask_dataframe() requires dask, not "
                       "available in this offline build")


def from_ray_dataset(*a, **k):
    raise RuntimeError("from_ray_dataset() requires ray; use "
                       "from_pydict/from_arrow")


def _gated_media(name, needs):
    def make(*a, **k):
        raise RuntimeError(f"{name} requires {needs}, not available in "
                           f"this offline build")
    make.__name__ = name
    return make


AudioFile = _gated_media("AudioFile", "an audio backend")
VideoFile = _gated_media("VideoFile", "ffmpeg")
Hdf5File = _gated_media("Hdf5File", "h5py")


def ImageFile(path):
    from .file import File
    return File(path)
