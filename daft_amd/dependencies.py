"""Lazy optional-dependency handles (ref: daft/dependencies.py): each
attribute imports on first use and raises a clear error if the package
is not installed in this offline image."""
import importlib


class _Lazy:
    def __init__(self, modname, alias):
        self._modname = modname
        self._alias = alias
        self._mod = None

    def _load(self):
        if self._mod is None:
            try:
                self._mod = importlib.import_module(self._modname)
            except ImportError as e:
                raise ImportError(
                    f"optional dependency {self._modname!r} "
                    f"({self._alias}) is not installed in this offline "
                    f"image") from e
        return self._mod

    def __getattr__(self, item):
        return getattr(self._load(), item)

    def module_available(self) -> bool:
        try:
            self._load()
            return True
        except ImportError:
            return False


np = _Lazy("numpy", "np")
pa = _Lazy("pyarrow", "pa")
pacsv = _Lazy("pyarrow.csv", "pacsv")
pads = _Lazy("pyarrow.dataset", "pads")
pafs = _Lazy("pyarrow.fs", "pafs")
pq = _Lazy("pyarrow.parquet", "pq")
fsspec = _Lazy("fsspec", "fsspec")
av = _Lazy("av", "av")
h5py = _Lazy("h5py", "h5py")
librosa = _Lazy("librosa", "librosa")
mcap = _Lazy("mcap", "mcap")
flight = _Lazy("pyarrow.flight", "flight")
confluent_kafka = _Lazy("confluent_kafka", "confluent_kafka")
