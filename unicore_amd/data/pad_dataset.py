"""Padding collation wrappers (parity: reference unicore/data/pad_dataset.py:12-38)."""

from . import data_utils
from .base_wrapper_dataset import BaseWrapperDataset


class PadDataset(BaseWrapperDataset):
    def __init__(self, dataset, pad_idx, left_pad, pad_to_multiple=8):
        super().__init__(dataset)
        self.pad_idx = pad_idx
        self.left_pad = left_pad
        self.pad_to_multiple = pad_to_multiple

    def collater(self, samples):
        return data_utils.collate_tokens(
            samples,
            self.pad_idx,
            left_pad=self.left_pad,
            pad_to_multiple=self.pad_to_multiple,
        )


class LeftPadDataset(PadDataset):
    def __init__(self, dataset, pad_idx, pad_to_multiple=8):
        super().__init__(dataset, pad_idx, left_pad=True, pad_to_multiple=pad_to_multiple)


class RightPadDataset(PadDataset):
    def __init__(self, dataset, pad_idx, pad_to_multiple=8):
        super().__init__(dataset, pad_idx, left_pad=False, pad_to_multiple=pad_to_multiple)


class RightPadDataset2D(BaseWrapperDataset):
    def __init__(self, dataset, pad_idx, left_pad=False, pad_to_multiple=8):
        super().__init__(dataset)
        self.pad_idx = pad_idx
        self.left_pad = left_pad
        self.pad_to_multiple = pad_to_multiple

    def collater(self, samples):
        return data_utils.collate_tokens_2d(
            samples,
            self.pad_idx,
            left_pad=self.left_pad,
            pad_to_multiple=self.pad_to_multiple,
        )


class RightPadDatasetCoord(BaseWrapperDataset):
    """Pad a list of (L_i, D) tensors along dim 0 to (B, L_max, D) — used for
    per-atom coordinate / feature arrays (Uni-Mol-style tasks)."""

    def __init__(self, dataset, pad_idx=0, pad_to_multiple=8):
        super().__init__(dataset)
        self.pad_idx = pad_idx
        self.pad_to_multiple = pad_to_multiple

    def collater(self, samples):
        size = max(v.size(0) for v in samples)
        if self.pad_to_multiple != 1 and size % self.pad_to_multiple != 0:
            size = int(
                ((size - 0.1) // self.pad_to_multiple + 1) * self.pad_to_multiple
            )
        d = samples[0].size(1)
        res = samples[0].new(len(samples), size, d).fill_(self.pad_idx)
        for i, v in enumerate(samples):
            res[i, : v.size(0)].copy_(v)
        return res
