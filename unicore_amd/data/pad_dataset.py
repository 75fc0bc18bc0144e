"""Padding collation wrappers (parity: reference
unicore/data/pad_dataset.py:12-38). All default to pad_to_multiple=8 so
collated shapes stay friendly to the fused kernels and GEMMs."""

from . import data_utils
from .base_wrapper_dataset import BaseWrapperDataset


class PadDataset(BaseWrapperDataset):
    """Collates 1-D token rows into a padded (B, L) batch."""

    def __init__(self, dataset, pad_idx, left_pad, pad_to_multiple=8):
        super().__init__(dataset=dataset)
        self.pad_idx, self.left_pad = pad_idx, left_pad
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
        super().__init__(dataset, pad_idx, True, pad_to_multiple)


class RightPadDataset(PadDataset):
    def __init__(self, dataset, pad_idx, pad_to_multiple=8):
        super().__init__(dataset, pad_idx, False, pad_to_multiple)


class RightPadDataset2D(BaseWrapperDataset):
    """Collates square (L, L) per-sample matrices into (B, L_max, L_max)."""

    def __init__(self, dataset, pad_idx, left_pad=False, pad_to_multiple=8):
        super().__init__(dataset=dataset)
        self.pad_idx, self.left_pad = pad_idx, left_pad
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
        super().__init__(dataset=dataset)
        self.pad_idx, self.pad_to_multiple = pad_idx, pad_to_multiple

    def collater(self, samples):
        from .data_utils import _round_up

        width = _round_up(max(v.size(0) for v in samples), self.pad_to_multiple)
        depth = samples[0].size(1)
        out = samples[0].new(len(samples), width, depth).fill_(self.pad_idx)
        for row, v in zip(out, samples):
            row[: v.size(0)].copy_(v)
        return out
