"""Tokenization wrapper datasets.

Parity: reference unicore/data/tokenize_dataset.py:13 (dictionary-based) and
unicore/data/bert_tokenize_dataset.py:14-37 (HuggingFace WordPiece).
"""

import logging

import torch

from functools import lru_cache

from .base_wrapper_dataset import BaseWrapperDataset
from .dictionary import Dictionary

logger = logging.getLogger(__name__)


class TokenizeDataset(BaseWrapperDataset):
    def __init__(
        self,
        dataset,
        dictionary: Dictionary,
        max_seq_len: int = 512,
    ):
        self.dataset = dataset
        self.dictionary = dictionary
        self.max_seq_len = max_seq_len

    @lru_cache(maxsize=16)
    def __getitem__(self, index: int):
        raw_data = self.dataset[index]
        assert len(raw_data) < self.max_seq_len and len(raw_data) > 0
        return self.dictionary.vec_index(raw_data).long()


class BertTokenizeDataset(BaseWrapperDataset):
    def __init__(
        self,
        dataset,
        dict_path: str,
        max_seq_len: int = 512,
    ):
        self.dataset = dataset
        self.dict_path = dict_path
        self.max_seq_len = max_seq_len
        self.tokenizer = None

    def _build_tokenizer(self):
        from tokenizers import BertWordPieceTokenizer

        return BertWordPieceTokenizer(self.dict_path, lowercase=True)

    def __getitem__(self, index: int):
        if self.tokenizer is None:
            # lazy per-worker construction (tokenizer is not picklable)
            self.tokenizer = self._build_tokenizer()
        raw_str = self.dataset[index]
        raw_str = raw_str.replace("<unk>", "[UNK]")
        output = self.tokenizer.encode(raw_str)
        ret = torch.LongTensor(output.ids[: self.max_seq_len])
        return ret
