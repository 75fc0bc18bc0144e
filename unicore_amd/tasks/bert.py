"""BERT masked-LM task over LMDB text data.

Functional parity with reference examples/bert/task.py:31-124, built in here
instead of requiring --user-dir. Pipeline per split:
LMDB -> WordPiece tokenize -> 80/10/10 mask -> right-pad -> nested dict ->
seeded shuffle order.
"""

import logging
import os

import numpy as np

from unicore_amd.data import (
    BertTokenizeDataset,
    Dictionary,
    LMDBDataset,
    MaskTokensDataset,
    NestedDictionaryDataset,
    RightPadDataset,
    SortDataset,
    data_utils,
)
from unicore_amd.tasks import UnicoreTask, register_task

logger = logging.getLogger(__name__)


@register_task("bert")
class BertTask(UnicoreTask):
    """Masked-language-model pretraining (BERT-style)."""

    @staticmethod
    def add_args(parser):
        parser.add_argument("data",
                            help="directory holding {split}.lmdb + dict.txt")
        parser.add_argument("--mask-prob", default=0.15, type=float,
                            help="fraction of tokens selected for masking")
        parser.add_argument("--leave-unmasked-prob", default=0.1, type=float,
                            help="share of selected tokens left unchanged")
        parser.add_argument("--random-token-prob", default=0.1, type=float,
                            help="share of selected tokens replaced randomly")

    def __init__(self, args, dictionary):
        super().__init__(args)
        self.dictionary, self.seed = dictionary, args.seed
        self.mask_idx = dictionary.add_symbol(
            "[MASK]", is_special=True
        )

    @classmethod
    def setup_task(cls, args, **kwargs):
        vocab = Dictionary.load(os.path.join(args.data, "dict.txt"))
        logger.info(f"dictionary: {len(vocab)} types")
        return cls(args, vocab)

    def _open_split(self, data_dir, split):
        """{split}.lmdb (reference format) or {split}.kv (our lmdb-free
        single-file store) — whichever exists."""
        lmdb_path = os.path.join(data_dir, f"{split}.lmdb")
        if os.path.isfile(lmdb_path):
            return LMDBDataset(lmdb_path)
        kv_path = os.path.join(data_dir, f"{split}.kv")
        if os.path.isfile(kv_path):
            from unicore_amd.data.kv_dataset import KVDataset

            return KVDataset(kv_path)
        raise FileNotFoundError(
            f"no {split}.lmdb or {split}.kv under {data_dir} "
            "(run examples/bert/prepare_corpus.py)"
        )

    def load_dataset(self, split, combine=False, **kwargs):
        """Build the {split} pipeline and register it under *split*."""
        data_dir = self.args.data
        tokens = BertTokenizeDataset(
            self._open_split(data_dir, split),
            os.path.join(data_dir, "dict.txt"),
            max_seq_len=self.args.max_seq_len,
        )

        cfg = self.args
        src, tgt = MaskTokensDataset.apply_mask(
            tokens, self.dictionary,
            pad_idx=self.dictionary.pad(), mask_idx=self.mask_idx,
            seed=cfg.seed, mask_prob=cfg.mask_prob,
            leave_unmasked_prob=cfg.leave_unmasked_prob,
            random_token_prob=cfg.random_token_prob,
        )

        with data_utils.numpy_seed(cfg.seed):
            order = np.random.permutation(len(src))

        pad = self.dictionary.pad()
        nested = NestedDictionaryDataset({
            "net_input": {
                "src_tokens": RightPadDataset(src, pad_idx=pad),
            },
            "target": RightPadDataset(tgt, pad_idx=pad),
        })
        self.datasets[split] = SortDataset(nested, sort_order=[order])
