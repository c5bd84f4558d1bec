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
