"""Shared test fixtures (reference tests/utils/utils.py:9-28)."""
import torch


class EchoDataset(torch.utils.data.Dataset):
    """Repeats one constant batch: deterministic shapes."""

    def __init__(self, data, repeat_count=100):
        self.data = data
        self.repeat_count = repeat_count

    def __len__(self):
        return self.repeat_count

    def __getitem__(self, _):
        return self.data


class RawDataset(torch.utils.data.Dataset):
    def __init__(self, items):
        self.items = items

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        return self.items[i]


def set_seed(seed=42):
    import random
    random.seed(seed)
    torch.manual_seed(seed)
