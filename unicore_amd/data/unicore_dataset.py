"""Dataset base class (parity: reference unicore/data/unicore_dataset.py:35-91)."""

import numpy as np
import torch.utils.data

from . import data_utils


class EpochListening:
    """Mixin: datasets that want to hear about epoch transitions."""

    @property
    def can_reuse_epoch_itr_across_epochs(self):
        """False when item SIZES can change between epochs (forces batch
        regeneration each epoch); datasets that depend on ``set_epoch`` for
        anything size-affecting should return False."""
        return True

    def set_epoch(self, epoch):
        """Called with the new (1-based) epoch number before it starts."""


class UnicoreDataset(torch.utils.data.Dataset, EpochListening):
    """torch Dataset + the batching/collation protocol the task layer uses."""

    def __getitem__(self, index):
        raise NotImplementedError("datasets implement item access")

    def __len__(self) -> int:
        raise NotImplementedError("datasets implement len()")

    def collater(self, samples):
        """Merge a list of samples into a model-ready mini-batch dict."""
        raise NotImplementedError("datasets implement collation")

    def ordered_indices(self):
        """Index order used for batch construction (identity by default)."""
        return np.arange(len(self), dtype=np.int64)

    @property
    def supports_prefetch(self):
        """True when :meth:`prefetch` is usable."""
        return False

    def attr(self, name, index):
        """Per-index attribute lookup hook (class attribute by default)."""
        return getattr(self, name, None)

    def prefetch(self, indices):
        """Bulk-load the given indices ahead of the epoch."""
        raise NotImplementedError("supports_prefetch datasets implement this")

    def batch_by_size(self, indices, batch_size=None,
                      required_batch_size_multiple=1):
        """Partition *indices* into batches of *batch_size* (rounded to the
        required multiple); see data_utils.batch_by_size."""
        return data_utils.batch_by_size(
            indices,
            batch_size=batch_size,
            required_batch_size_multiple=required_batch_size_multiple,
        )
