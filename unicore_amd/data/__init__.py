"""Data pipeline: composable wrapper datasets + resumable iterators
(parity: reference unicore/data/__init__.py:9-33)."""

from .unicore_dataset import EpochListening, UnicoreDataset
from .base_wrapper_dataset import BaseWrapperDataset
from .dictionary import Dictionary
from .lru_cache_dataset import LRUCacheDataset
from .lmdb_dataset import LMDBDataset
from .tokenize_dataset import BertTokenizeDataset, TokenizeDataset
from .mask_tokens_dataset import MaskTokensDataset
from .pad_dataset import (
    LeftPadDataset,
    PadDataset,
    RightPadDataset,
    RightPadDataset2D,
    RightPadDatasetCoord,
)
from .sort_dataset import EpochShuffleDataset, SortDataset
from .nested_dictionary_dataset import NestedDictionaryDataset
from .misc_datasets import (
    AppendTokenDataset,
    FromNumpyDataset,
    NumelDataset,
    NumSamplesDataset,
    PrependTokenDataset,
    RawArrayDataset,
    RawLabelDataset,
    RawNumpyDataset,
)
from . import data_utils, iterators
from .iterators import (
    BufferedIterator,
    CountingIterator,
    EpochBatchIterator,
    GroupedIterator,
    ShardedIterator,
)

__all__ = [
    "BufferedIterator",
    "CountingIterator",
    "EpochBatchIterator",
    "GroupedIterator",
    "ShardedIterator",
    "AppendTokenDataset",
    "BaseWrapperDataset",
    "BertTokenizeDataset",
    "Dictionary",
    "EpochListening",
    "EpochShuffleDataset",
    "FromNumpyDataset",
    "LeftPadDataset",
    "LMDBDataset",
    "LRUCacheDataset",
    "MaskTokensDataset",
    "NestedDictionaryDataset",
    "NumelDataset",
    "NumSamplesDataset",
    "PadDataset",
    "PrependTokenDataset",
    "RawArrayDataset",
    "RawLabelDataset",
    "RawNumpyDataset",
    "RightPadDataset",
    "RightPadDataset2D",
    "RightPadDatasetCoord",
    "SortDataset",
    "TokenizeDataset",
    "UnicoreDataset",
    "data_utils",
    "iterators",
]
