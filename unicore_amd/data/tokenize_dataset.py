"""Tokenization wrappers.

Parity: reference unicore/data/tokenize_dataset.py:13 (dictionary-based) and
unicore/data/bert_tokenize_dataset.py:14-37 (HuggingFace WordPiece).
"""

import logging
from functools import lru_cache

import torch

from .base_wrapper_dataset import BaseWrapperDataset
from .dictionary import Dictionary

logger = logging.getLogger(__name__)


class TokenizeDataset(BaseWrapperDataset):
    """Maps pre-split symbol sequences to index tensors via a Dictionary."""

    def __init__(self, dataset, dictionary: Dictionary, max_seq_len=512):
        self.dataset = dataset
        self.dictionary = dictionary
        self.max_seq_len = max_seq_len

    @lru_cache(16)
    def __getitem__(self, index):
        symbols = self.dataset[index]
        assert 0 < len(symbols) < self.max_seq_len
        return self.dictionary.vec_index(symbols).long()


class BertTokenizeDataset(BaseWrapperDataset):
    """Runs HuggingFace BertWordPieceTokenizer over raw text lines.

    The tokenizer is built lazily in each DataLoader worker (it does not
    pickle across the fork).
    """

    def __init__(self, dataset, dict_path: str, max_seq_len=512):
        self.dataset = dataset
        self.dict_path = dict_path
        self.max_seq_len = max_seq_len
        self.tokenizer = None

    def _build_tokenizer(self):
        from tokenizers import BertWordPieceTokenizer

        return BertWordPieceTokenizer(self.dict_path, lowercase=True)

    def __getitem__(self, index):
        if self.tokenizer is None:
            self.tokenizer = self._build_tokenizer()
        text = self.dataset[index].replace("<unk>", "[UNK]")
        encoded = self.tokenizer.encode(text)
        return torch.LongTensor(encoded.ids[: self.max_seq_len])
