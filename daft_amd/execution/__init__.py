from .runner import NativeRunner  # noqa: F401
