"""Index-ordering wrappers (parity: reference
unicore/data/sort_dataset.py:12-42)."""

import numpy as np

from .base_wrapper_dataset import BaseWrapperDataset
from .data_utils import numpy_seed


class SortDataset(BaseWrapperDataset):
    """Orders indices by one or more sort keys (last key is primary,
    np.lexsort semantics) — used to group similar-length items."""

    def __init__(self, dataset, sort_order):
        super().__init__(dataset=dataset)
        self.sort_keys = (
            sort_order if isinstance(sort_order, (list, tuple)) else [sort_order]
        )
        assert all(len(key) == len(dataset) for key in self.sort_keys)

    def ordered_indices(self):
        return np.lexsort(tuple(self.sort_keys))


class EpochShuffleDataset(BaseWrapperDataset):
    """Deterministic fresh permutation per epoch (seeded by seed+epoch)."""

    def __init__(self, dataset, size, seed):
        super().__init__(dataset=dataset)
        self.size, self.seed = size, seed
        self.set_epoch(1)

    def set_epoch(self, epoch, **unused):
        super().set_epoch(epoch)
        with numpy_seed(self.seed + epoch - 1):
            self.permutation = np.random.permutation(self.size)

    def ordered_indices(self):
        return self.permutation

    @property
    def can_reuse_epoch_itr_across_epochs(self):
        return False  # the permutation is epoch-dependent
