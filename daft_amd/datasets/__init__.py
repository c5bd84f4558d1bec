"""Dataset loaders (capability of /root/reference/daft/datasets/:
common_crawl, droid, lerobot).  This environment has no egress, so every
loader resolves against a LOCAL MIRROR root (`data_root`) laid out like
the upstream bucket; the manifest/path resolution logic matches the
reference's."""
from .common_crawl import common_crawl
from . import lerobot

__all__ = ["common_crawl", "lerobot"]
