"""BERT-style 80/10/10 masking wrapper (parity: reference
unicore/data/mask_tokens_dataset.py:16-132).

Noise is drawn under ``numpy_seed(seed, epoch, index)``, which makes every
item deterministic per (epoch, index) AND keeps the source/target dataset
pair (two instances over one shared base) in agreement on the mask
positions: both consume the identical RNG stream up to the point where the
target variant returns.
"""

from functools import lru_cache

import numpy as np
import torch

from .data_utils import numpy_seed
from .base_wrapper_dataset import BaseWrapperDataset
from .dictionary import Dictionary
from .lru_cache_dataset import LRUCacheDataset


class MaskTokensDataset(BaseWrapperDataset):
    """Masked-LM noising wrapper.

    One instance yields the corrupted inputs, a twin instance
    (``return_masked_tokens=True``) yields the targets: original tokens at
    masked positions, pad everywhere else. Build both through
    :meth:`apply_mask` so they share one cached base dataset.
    """

    @classmethod
    def apply_mask(cls, dataset: torch.utils.data.Dataset, *args, **kwargs):
        """Return the (source, target) dataset pair for masked-LM training."""
        shared = LRUCacheDataset(dataset)
        make = lambda as_target: LRUCacheDataset(
            cls(shared, *args, **kwargs, return_masked_tokens=as_target)
        )
        return make(False), make(True)

    def __init__(
        self,
        dataset,
        vocab,
        pad_idx,
        mask_idx,
        return_masked_tokens=False,
        seed=1,
        mask_prob=0.15,
        leave_unmasked_prob=0.1,
        random_token_prob=0.1,
    ):
        assert 0.0 < mask_prob < 1.0, "mask_prob must be in (0, 1)"
        assert 0.0 <= random_token_prob <= 1.0 >= leave_unmasked_prob >= 0.0
        assert random_token_prob + leave_unmasked_prob <= 1.0, \
            "the random/unmasked splits cannot exceed the whole mask"

        self.dataset, self.vocab, self.seed = dataset, vocab, seed
        self.pad_idx, self.mask_idx = pad_idx, mask_idx
        self.return_masked_tokens = bool(return_masked_tokens)
        self.mask_prob = float(mask_prob)
        self.leave_unmasked_prob = float(leave_unmasked_prob)
        self.random_token_prob = float(random_token_prob)
        self.epoch = None

        if random_token_prob > 0.0:
            # uniform over the non-special vocabulary
            w = np.ones(len(vocab))
            w[vocab.special_index()] = 0
            self.weights = w / w.sum()

    @property
    def can_reuse_epoch_itr_across_epochs(self):
        # only the noise changes across epochs, never the item sizes
        return True

    def set_epoch(self, epoch, **unused):
        super().set_epoch(epoch)
        self.epoch = epoch  # keyed into the per-item RNG seed

    def __getitem__(self, index):
        return self.__getitem_cached__(self.epoch, index)

    @lru_cache(16)
    def __getitem_cached__(self, epoch, index):
        with numpy_seed(self.seed, epoch, index):
            tokens = self.dataset[index]
            sz = tokens.shape[0] if hasattr(tokens, 'shape') else len(tokens)
            assert self.mask_idx not in tokens, (
                f"Dataset contains mask_idx (={self.mask_idx}), "
                "this is not expected!"
            )

            # choose the mask; np.random.rand() makes E[num_mask] exact
            # despite int truncation
            picked = np.random.choice(
                sz, int(self.mask_prob * sz + np.random.rand()), replace=False
            )
            mask = np.zeros(sz, dtype=bool)
            mask[picked[picked < sz]] = True

            if self.return_masked_tokens:
                # target variant: originals at masked slots, pad elsewhere.
                # Returns here so both variants consumed the same RNG calls.
                target = np.full(sz, self.pad_idx)
                sel = torch.from_numpy(mask.astype(np.uint8)) == 1
                target[mask] = tokens[sel]
                return torch.from_numpy(target)

            # split the masked positions into keep-original / random-token
            # subsets per the 80/10/10 recipe
            unmask, rand_mask = None, None
            either_prob = self.random_token_prob + self.leave_unmasked_prob
            if either_prob > 0.0:
                either = mask & (np.random.rand(sz) < either_prob)
                if self.random_token_prob == 0.0:
                    unmask = either
                elif self.leave_unmasked_prob == 0.0:
                    rand_mask = either
                else:
                    keep_original = np.random.rand(sz) < (
                        self.leave_unmasked_prob / either_prob
                    )
                    unmask = either & keep_original
                    rand_mask = either & ~keep_original

            if unmask is not None:
                mask ^= unmask

            corrupted = np.copy(tokens)
            corrupted[mask] = self.mask_idx
            if rand_mask is not None and rand_mask.sum() > 0:
                corrupted[rand_mask] = np.random.choice(
                    len(self.vocab), rand_mask.sum(), p=self.weights
                )
            return torch.from_numpy(corrupted)
