from .runner import DistributedRunner  # noqa: F401
