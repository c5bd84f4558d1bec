from .translate import translate  # noqa: F401
