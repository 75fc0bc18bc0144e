"""Small LRU cache over item fetch + collation (parity: reference
unicore/data/lru_cache_dataset.py:12).

Used around MaskTokensDataset pairs so the shared base item is loaded once
per (source, target) access.
"""

from functools import lru_cache

from .base_wrapper_dataset import BaseWrapperDataset


class LRUCacheDataset(BaseWrapperDataset):
    def __init__(self, dataset, token=None):
        super().__init__(dataset=dataset)

    @lru_cache(16)
    def __getitem__(self, index):
        return super().__getitem__(index)

    @lru_cache(16)
    def collater(self, samples):
        return super().collater(samples)
