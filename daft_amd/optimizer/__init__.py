from .optimizer import optimize  # noqa: F401
