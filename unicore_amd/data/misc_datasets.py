"""Small wrapper datasets.

Parity: reference unicore/data/{numel_dataset.py:13, num_samples_dataset.py:10,
prepend_token_dataset.py:14, append_token_dataset.py:14, raw_dataset.py,
from_numpy_dataset.py:11}.
"""

import numpy as np
import torch

from functools import lru_cache

from .base_wrapper_dataset import BaseWrapperDataset
from .unicore_dataset import UnicoreDataset


def _as_tensor(item):
    return item if torch.is_tensor(item) else torch.from_numpy(np.asarray(item))


class NumelDataset(BaseWrapperDataset):
    """Element count of each item; collates to a sum or a count vector."""

    def __init__(self, dataset, reduce=False):
        super().__init__(dataset=dataset)
        self.reduce = reduce

    def __getitem__(self, index):
        item = self.dataset[index]
        return torch.numel(item) if torch.is_tensor(item) else np.size(item)

    def __len__(self) -> int:
        return len(self.dataset)

    def collater(self, samples):
        return sum(samples) if self.reduce else torch.tensor(samples)


class NumSamplesDataset(UnicoreDataset):
    """Constant 1 per item; collates to the batch row count."""

    def __getitem__(self, index):
        return 1

    def __len__(self) -> int:
        return 0

    def collater(self, samples):
        return sum(samples)


class PrependTokenDataset(BaseWrapperDataset):
    def __init__(self, dataset, token=None):
        super().__init__(dataset=dataset)
        self.token = token

    @lru_cache(16)
    def __getitem__(self, idx):
        item = self.dataset[idx]
        if self.token is None:
            return item
        return torch.cat([item.new([self.token]), item])


class AppendTokenDataset(BaseWrapperDataset):
    def __init__(self, dataset, token=None):
        super().__init__(dataset=dataset)
        self.token = token

    @lru_cache(16)
    def __getitem__(self, idx):
        item = self.dataset[idx]
        if self.token is None:
            return item
        return torch.cat([item, item.new([self.token])])


class RawLabelDataset(UnicoreDataset):
    """Plain python list of labels as a dataset."""

    def __init__(self, labels):
        super().__init__()
        self.values = labels

    def __getitem__(self, index):
        return self.values[index]

    def __len__(self) -> int:
        return len(self.values)

    def collater(self, samples):
        return torch.tensor(samples)


class RawArrayDataset(BaseWrapperDataset):
    """Identity wrapper (items pass through; collation delegates)."""


class RawNumpyDataset(BaseWrapperDataset):
    """Items coerced to tensors on access."""

    def __getitem__(self, index):
        return _as_tensor(self.dataset[index])


class FromNumpyDataset(BaseWrapperDataset):
    @lru_cache(16)
    def __getitem__(self, idx):
        return _as_tensor(self.dataset[idx])
