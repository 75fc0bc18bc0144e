"""Synthetic masked-MSA pretraining task for the Evoformer stress config
(BASELINE.json config 5): random MSAs (S sequences x L residues), 15%
masking, masked-residue prediction."""

import logging

import numpy as np
import torch

from unicore_amd.data import Dictionary, UnicoreDataset, data_utils
from unicore_amd.tasks import UnicoreTask, register_task

logger = logging.getLogger(__name__)


def make_residue_dictionary(n_types=26):
    d = Dictionary()
    for sym in ("[CLS]", "[PAD]", "[SEP]", "[UNK]"):
        d.add_symbol(sym, is_special=True)
    for i in range(n_types):
        d.add_symbol(f"res{i}")
    return d


class SyntheticMSADataset(UnicoreDataset):
    def __init__(self, size, n_seq, seq_len, dictionary, mask_idx, seed,
                 mask_prob=0.15):
        super().__init__()
        self.size = size
        self.n_seq = n_seq
        self.seq_len = seq_len
        self.dictionary = dictionary
        self.mask_idx = mask_idx
        self.seed = seed
        self.mask_prob = mask_prob
        self.epoch = 1

    def set_epoch(self, epoch, **unused):
        self.epoch = epoch

    def __len__(self):
        return self.size

    def __getitem__(self, index):
        with data_utils.numpy_seed(self.seed, self.epoch, index):
            toks = np.random.randint(
                5, self.mask_idx, size=(self.n_seq, self.seq_len)
            )
            mask = np.random.rand(self.n_seq, self.seq_len) < self.mask_prob
        target = np.full_like(toks, self.dictionary.pad())
        target[mask] = toks[mask]
        src = toks.copy()
        src[mask] = self.mask_idx
        return {
            "src_tokens": torch.from_numpy(src.astype(np.int64)),
            "target": torch.from_numpy(target.astype(np.int64)),
        }


class _StackDataset(UnicoreDataset):
    """Fixed-shape samples -> stacked batch tensors."""

    def __init__(self, base, key):
        super().__init__()
        self.base = base
        self.key = key

    def __len__(self):
        return len(self.base)

    def __getitem__(self, index):
        return self.base[index][self.key]

    def collater(self, samples):
        return torch.stack(samples)

    def set_epoch(self, epoch, **unused):
        self.base.set_epoch(epoch)


@register_task("evoformer_synthetic")
class EvoformerSyntheticTask(UnicoreTask):
    @staticmethod
    def add_args(parser):
        parser.add_argument("--dataset-size", default=64, type=int)
        parser.add_argument("--msa-depth", default=32, type=int)
        parser.add_argument("--residues", default=64, type=int)
        parser.add_argument("--mask-prob", default=0.15, type=float)

    def __init__(self, args, dictionary):
        super().__init__(args)
        self.dictionary = dictionary
        self.seed = args.seed
        self.mask_idx = dictionary.add_symbol("[MASK]", is_special=True)
        dictionary.pad_to_multiple_(64)

    @classmethod
    def setup_task(cls, args, **kwargs):
        d = make_residue_dictionary()
        logger.info("residue dictionary: {} types".format(len(d)))
        return cls(args, d)

    def load_dataset(self, split, combine=False, **kwargs):
        from unicore_amd.data import NestedDictionaryDataset

        base = SyntheticMSADataset(
            size=self.args.dataset_size,
            n_seq=self.args.msa_depth,
            seq_len=self.args.residues,
            dictionary=self.dictionary,
            mask_idx=self.mask_idx,
            seed=self.seed + (0 if split == "train" else 1),
            mask_prob=self.args.mask_prob,
        )
        self.datasets[split] = NestedDictionaryDataset(
            {
                "net_input": {"src_tokens": _StackDataset(base, "src_tokens")},
                "target": _StackDataset(base, "target"),
            }
        )
