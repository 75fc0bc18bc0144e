"""Synthetic 3D-molecule pretraining task (BASELINE.json stress config 4).

Random atom types + 3D conformations; 15% of atoms are masked (token target)
and gaussian noise is added to every coordinate (denoising target).  Batches
carry src_tokens, src_coord, target tokens and coord_target (= -noise)."""

import logging

import numpy as np
import torch

from unicore_amd.data import (
    Dictionary,
    NestedDictionaryDataset,
    RightPadDataset,
    RightPadDatasetCoord,
    SortDataset,
    UnicoreDataset,
    data_utils,
)
from unicore_amd.tasks import UnicoreTask, register_task

logger = logging.getLogger(__name__)


def make_atom_dictionary(n_types=32):
    d = Dictionary()
    for sym in ("[CLS]", "[PAD]", "[SEP]", "[UNK]"):
        d.add_symbol(sym, is_special=True)
    for i in range(n_types):
        d.add_symbol(f"atom{i}")
    return d


class SyntheticMolDataset(UnicoreDataset):
    """Deterministic random molecules: tokens, noised coords, targets."""

    def __init__(self, size, n_atoms, dictionary, mask_idx, seed, mask_prob=0.15,
                 noise_std=0.3):
        super().__init__()
        self.size = size
        self.n_atoms = n_atoms
        self.dictionary = dictionary
        self.mask_idx = mask_idx
        self.seed = seed
        self.mask_prob = mask_prob
        self.noise_std = noise_std
        self.epoch = 1

    def set_epoch(self, epoch, **unused):
        self.epoch = epoch

    def __len__(self):
        return self.size

    def __getitem__(self, index):
        with data_utils.numpy_seed(self.seed, self.epoch, index):
            n = self.n_atoms
            toks = np.random.randint(5, self.mask_idx, size=n)
            coord = np.random.randn(n, 3).astype(np.float32) * 2.0
            noise = (np.random.randn(n, 3).astype(np.float32) * self.noise_std)
            mask = np.random.rand(n) < self.mask_prob
        target = np.full(n, self.dictionary.pad(), dtype=np.int64)
        target[mask] = toks[mask]
        src = toks.copy()
        src[mask] = self.mask_idx
        return {
            "src_tokens": torch.from_numpy(src),
            "src_coord": torch.from_numpy(coord + noise),
            "target": torch.from_numpy(target),
            "coord_target": torch.from_numpy(-noise),
        }


class _Field(UnicoreDataset):
    def __init__(self, base, key):
        super().__init__()
        self.base = base
        self.key = key

    def __len__(self):
        return len(self.base)

    def __getitem__(self, index):
        return self.base[index][self.key]

    def set_epoch(self, epoch, **unused):
        self.base.set_epoch(epoch)


@register_task("unimol_synthetic")
class UniMolSyntheticTask(UnicoreTask):
    @staticmethod
    def add_args(parser):
        parser.add_argument("--dataset-size", default=256, type=int)
        parser.add_argument("--atoms-per-mol", default=128, type=int)
        parser.add_argument("--atom-types", default=32, type=int)
        parser.add_argument("--mask-prob", default=0.15, type=float)
        parser.add_argument("--coord-noise-std", default=0.3, type=float)
        parser.add_argument("--coord-loss-weight", default=1.0, type=float)

    def __init__(self, args, dictionary):
        super().__init__(args)
        self.dictionary = dictionary
        self.seed = args.seed
        self.mask_idx = dictionary.add_symbol("[MASK]", is_special=True)
        dictionary.pad_to_multiple_(64)

    @classmethod
    def setup_task(cls, args, **kwargs):
        d = make_atom_dictionary(getattr(args, "atom_types", 32))
        logger.info("atom dictionary: {} types".format(len(d)))
        return cls(args, d)

    def load_dataset(self, split, combine=False, **kwargs):
        base = SyntheticMolDataset(
            size=self.args.dataset_size,
            n_atoms=self.args.atoms_per_mol,
            dictionary=self.dictionary,
            mask_idx=self.mask_idx,
            seed=self.seed + (0 if split == "train" else 1),
            mask_prob=self.args.mask_prob,
            noise_std=self.args.coord_noise_std,
        )
        pad = self.dictionary.pad()
        ds = NestedDictionaryDataset(
            {
                "net_input": {
                    "src_tokens": RightPadDataset(_Field(base, "src_tokens"),
                                                  pad_idx=pad),
                    "src_coord": RightPadDatasetCoord(_Field(base, "src_coord"),
                                                   pad_idx=0),
                },
                "target": RightPadDataset(_Field(base, "target"), pad_idx=pad),
                "coord_target": RightPadDatasetCoord(_Field(base, "coord_target"),
                                                  pad_idx=0),
            }
        )
        with data_utils.numpy_seed(self.seed):
            shuffle = np.random.permutation(len(ds))
        self.datasets[split] = SortDataset(ds, sort_order=[shuffle])
