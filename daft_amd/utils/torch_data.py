"""torch dataset adapters (ref capability: daft's to_torch_map_dataset)."""
from __future__ import annotations

import torch


class DictDataset(torch.utils.data.Dataset):
    def __init__(self, data: dict):
        self.data = data
        self.names = list(data.keys())
        self.n = len(data[self.names[0]]) if self.names else 0

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        return {k: self.data[k][i] for k in self.names}


class IterDataset(torch.utils.data.IterableDataset):
    """Streams rows batch-by-batch from a lazy DataFrame (ref capability:
    daft/dataframe/to_torch.py DaftTorchIterableDataset).  Re-executes the
    plan on each iteration; honors torch DataLoader worker sharding by
    round-robin over worker id."""

    def __init__(self, df):
        self.df = df

    def __iter__(self):
        info = torch.utils.data.get_worker_info()
        it = self.df.iter_rows()
        if info is None or info.num_workers <= 1:
            yield from it
            return
        for i, row in enumerate(it):
            if i % info.num_workers == info.id:
                yield row
