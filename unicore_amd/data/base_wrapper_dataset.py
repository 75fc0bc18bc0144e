"""Delegating wrapper base for composable datasets (parity: reference
unicore/data/base_wrapper_dataset.py:12)."""

from torch.utils.data.dataloader import default_collate

from .unicore_dataset import UnicoreDataset


class BaseWrapperDataset(UnicoreDataset):
    """Forwards the whole UnicoreDataset protocol to ``self.dataset``;
    subclasses override just the piece they transform."""

    def __init__(self, dataset):
        super().__init__()
        self.dataset = dataset  # the wrapped inner dataset

    def __getitem__(self, index):
        return self.dataset[index]

    def __len__(self) -> int:
        return len(self.dataset)

    def collater(self, samples):
        inner = getattr(self.dataset, "collater", None)
        return inner(samples) if inner is not None else default_collate(samples)

    def ordered_indices(self):
        return self.dataset.ordered_indices()  # delegate ordering

    @property
    def supports_prefetch(self):
        return getattr(self.dataset, "supports_prefetch", False)

    def attr(self, name, index):
        return self.dataset.attr(name, index)

    def prefetch(self, indices) -> None:
        self.dataset.prefetch(indices)

    @property
    def can_reuse_epoch_itr_across_epochs(self):
        return self.dataset.can_reuse_epoch_itr_across_epochs  # delegate

    def set_epoch(self, epoch):
        super().set_epoch(epoch)
        inner = getattr(self.dataset, "set_epoch", None)
        if inner is not None:
            inner(epoch)
