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


class NumelDataset(BaseWrapperDataset):
    def __init__(self, dataset, reduce=False):
        super().__init__(dataset)
        self.reduce = reduce

    def __getitem__(self, index):
        item = self.dataset[index]
        if torch.is_tensor(item):
            return torch.numel(item)
        else:
            return np.size(item)

    def __len__(self):
        return len(self.dataset)

    def collater(self, samples):
        if self.reduce:
            return sum(samples)
        else:
            return torch.tensor(samples)


class NumSamplesDataset(UnicoreDataset):
    def __getitem__(self, index):
        return 1

    def __len__(self):
        return 0

    def collater(self, samples):
        return sum(samples)


class PrependTokenDataset(BaseWrapperDataset):
    def __init__(self, dataset, token=None):
        super().__init__(dataset)
        self.token = token

    @lru_cache(maxsize=16)
    def __getitem__(self, idx):
        item = self.dataset[idx]
        if self.token is not None:
            item = torch.cat([item.new([self.token]), item])
        return item


class AppendTokenDataset(BaseWrapperDataset):
    def __init__(self, dataset, token=None):
        super().__init__(dataset)
        self.token = token

    @lru_cache(maxsize=16)
    def __getitem__(self, idx):
        item = self.dataset[idx]
        if self.token is not None:
            item = torch.cat([item, item.new([self.token])])
        return item


class RawLabelDataset(UnicoreDataset):
    def __init__(self, labels):
        super().__init__()
        self.labels = labels

    def __getitem__(self, index):
        return self.labels[index]

    def __len__(self):
        return len(self.labels)

    def collater(self, samples):
        return torch.tensor(samples)


class RawArrayDataset(BaseWrapperDataset):
    def __init__(self, dataset):
        super().__init__(dataset)

    def __getitem__(self, index):
        return self.dataset[index]

    def collater(self, samples):
        if hasattr(self.dataset, "collater"):
            return self.dataset.collater(samples)
        else:
            return torch.utils.data.dataloader.default_collate(samples)


class RawNumpyDataset(BaseWrapperDataset):
    def __init__(self, dataset):
        super().__init__(dataset)

    def __getitem__(self, index):
        item = self.dataset[index]
        if not torch.is_tensor(item):
            item = torch.from_numpy(np.asarray(item))
        return item

    def collater(self, samples):
        if hasattr(self.dataset, "collater"):
            return self.dataset.collater(samples)
        else:
            return torch.utils.data.dataloader.default_collate(samples)


class FromNumpyDataset(BaseWrapperDataset):
    @lru_cache(maxsize=16)
    def __getitem__(self, idx):
        item = self.dataset[idx]
        if not torch.is_tensor(item):
            item = torch.from_numpy(np.asarray(item))
        return item
