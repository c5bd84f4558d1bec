"""LeRobot dataset loader (ref capability:
/root/reference/daft/datasets/lerobot.py — parse the dataset's
meta/info.json + episode parquet shards into one DataFrame).  Offline:
`root` is a local checkout of a LeRobot dataset repo."""
from __future__ import annotations

import glob
import json
import os


def load(root: str, episodes=None):
    """Read a LeRobot-format dataset directory: meta/info.json describes
    chunked `data/chunk-*/episode_*.parquet` shards; returns the
    concatenated DataFrame (optionally restricted to `episodes`)."""
    from .. import read_parquet
    info_path = os.path.join(root, "meta", "info.json")
    if not os.path.exists(info_path):
        raise FileNotFoundError(f"not a LeRobot dataset root: {info_path}")
    info = json.load(open(info_path))
    data_glob = info.get("data_path")
    if data_glob:
        # template like data/chunk-{episode_chunk:03d}/episode_{episode_index:06d}.parquet
        base = data_glob.split("{")[0]
        paths = sorted(glob.glob(os.path.join(root, base + "**", "*.parquet"),
                                 recursive=True))
    else:
        paths = sorted(glob.glob(os.path.join(root, "data", "**",
                                              "*.parquet"), recursive=True))
    if episodes is not None:
        keep = {f"episode_{e:06d}.parquet" for e in episodes}
        paths = [p for p in paths if os.path.basename(p) in keep]
    if not paths:
        raise FileNotFoundError(f"no parquet shards under {root}/data")
    return read_parquet(paths)


def info(root: str) -> dict:
    return json.load(open(os.path.join(root, "meta", "info.json")))
