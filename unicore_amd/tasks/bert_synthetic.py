"""Synthetic BERT masked-LM task: random tokens, no disk data.

Used by the CPU plumbing config (BASELINE.json config 1), the test suite and
bench.py: exercises the full pipeline (mask -> pad -> nest -> shuffle ->
trainer) on deterministic synthetic sequences, with no LMDB/network
dependency.
"""

import logging

import numpy as np
import torch

from unicore_amd.data import (
    Dictionary,
    FromNumpyDataset,
    MaskTokensDataset,
    NestedDictionaryDataset,
    RightPadDataset,
    SortDataset,
    UnicoreDataset,
    data_utils,
)
from unicore_amd.tasks import UnicoreTask, register_task

logger = logging.getLogger(__name__)


def make_synthetic_dictionary(vocab_size=30522):
    d = Dictionary()
    for sym in ("[CLS]", "[PAD]", "[SEP]", "[UNK]"):
        d.add_symbol(sym, is_special=True)
    for i in range(vocab_size - len(d)):
        d.add_symbol(f"tok{i}")
    return d


class SyntheticTokensDataset(UnicoreDataset):
    """Deterministic random token sequences (seeded by index)."""

    def __init__(self, size, seq_len, vocab_size, seed, num_special=5, fixed_len=True):
        super().__init__()
        self.size = size
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed
        self.num_special = num_special
        self.fixed_len = fixed_len

    def __len__(self):
        return self.size

    def __getitem__(self, index):
        with data_utils.numpy_seed(self.seed, index):
            if self.fixed_len:
                n = self.seq_len
            else:
                n = np.random.randint(self.seq_len // 2, self.seq_len + 1)
            toks = np.random.randint(self.num_special, self.vocab_size, size=n)
        return torch.from_numpy(toks.astype(np.int64))


@register_task("bert_synthetic")
class BertSyntheticTask(UnicoreTask):
    @staticmethod
    def add_args(parser):
        parser.add_argument("--vocab-size", default=30522, type=int)
        parser.add_argument("--dataset-size", default=512, type=int)
        parser.add_argument("--tokens-per-sample", default=512, type=int)
        parser.add_argument("--variable-seq-len", action="store_true")
        parser.add_argument("--mask-prob", default=0.15, type=float)
        parser.add_argument("--leave-unmasked-prob", default=0.1, type=float)
        parser.add_argument("--random-token-prob", default=0.1, type=float)

    def __init__(self, args, dictionary):
        super().__init__(args)
        self.dictionary, self.seed = dictionary, args.seed
        self.mask_idx = dictionary.add_symbol(
            "[MASK]", is_special=True
        )
        # vocab to a multiple of 64: the lm-head GEMM and fused cross
        # entropy want an even, vectorizable inner dimension
        dictionary.pad_to_multiple_(64)

    @classmethod
    def setup_task(cls, args, **kwargs):
        vocab = make_synthetic_dictionary(getattr(args, "vocab_size", 30522))
        logger.info(f"synthetic dictionary: {len(vocab)} types")
        return cls(args, vocab)

    def load_dataset(self, split, combine=False, **kwargs):
        seq_len = min(self.args.tokens_per_sample, self.args.max_seq_len - 1)
        raw = SyntheticTokensDataset(
            size=self.args.dataset_size,
            seq_len=seq_len,
            # sample strictly below the [MASK] symbol (filler symbols used
            # to pad the vocab to a multiple of 64 sit above it)
            vocab_size=self.mask_idx,
            seed=self.seed + (0 if split == "train" else 1),
            fixed_len=not getattr(self.args, "variable_seq_len", False),
        )
        dataset = FromNumpyDataset(raw)

        cfg = self.args
        src, tgt = MaskTokensDataset.apply_mask(
            dataset, self.dictionary,
            pad_idx=self.dictionary.pad(), mask_idx=self.mask_idx,
            seed=self.seed, mask_prob=cfg.mask_prob,
            leave_unmasked_prob=cfg.leave_unmasked_prob,
            random_token_prob=cfg.random_token_prob,
        )

        with data_utils.numpy_seed(self.seed):
            order = np.random.permutation(len(src))

        pad = self.dictionary.pad()
        nested = NestedDictionaryDataset({
            "net_input": {
                "src_tokens": RightPadDataset(src, pad_idx=pad),
            },
            "target": RightPadDataset(tgt, pad_idx=pad),
        })
        self.datasets[split] = SortDataset(nested, sort_order=[order])
