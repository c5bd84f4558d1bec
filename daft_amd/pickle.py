"""daft.pickle shim (ref: daft/pickle/__init__.py): cloudpickle-backed
dumps/loads used for shipping UDFs."""
try:
    import cloudpickle as _cp
except ImportError:                      # pragma: no cover
    import pickle as _cp

def dumps(obj) -> bytes:
    return _cp.dumps(obj)

def loads(data: bytes):
    return _cp.loads(data)
