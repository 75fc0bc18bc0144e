"""Collation, seeding and batching helpers (parity: reference
unicore/data/data_utils.py — collate_tokens:17, collate_dict:63,
numpy_seed:83, batch_by_size:107).

Both collate variants share one implementation here: the 2-D case pads a
square (len, len) block per sample instead of a row.
"""

import contextlib
import logging

import numpy as np
import torch

logger = logging.getLogger(__name__)


def _round_up(size, multiple):
    """Smallest value >= size that divides *multiple* (no-op for multiple=1)."""
    if multiple == 1 or size % multiple == 0:
        return size
    return int(((size - 0.1) // multiple + 1) * multiple)


def _collate_padded(values, pad_idx, left_pad, pad_to_length, pad_to_multiple,
                    square):
    width = max(v.size(0) for v in values)
    if pad_to_length is not None:
        width = max(width, pad_to_length)
    width = _round_up(width, pad_to_multiple)
    shape = (len(values), width, width) if square else (len(values), width)
    out = values[0].new(*shape).fill_(pad_idx)
    for row, v in zip(out, values):
        n = len(v)
        if square:
            dst = row[width - n:, width - n:] if left_pad else row[:n, :n]
        else:
            dst = row[width - n:] if left_pad else row[:n]
        assert dst.numel() == v.numel()
        dst.copy_(v)
    return out


def collate_tokens(values, pad_idx, left_pad=False, pad_to_length=None,
                   pad_to_multiple=1):
    """Pad a list of 1-D tensors into one (B, L) tensor
    (reference unicore/data/data_utils.py:17-38)."""
    return _collate_padded(values, pad_idx, left_pad, pad_to_length,
                           pad_to_multiple, square=False)


def collate_tokens_2d(values, pad_idx, left_pad=False, pad_to_length=None,
                      pad_to_multiple=1):
    """Pad a list of 2-D (L_i, L_i) tensors into one (B, L, L) tensor
    (reference unicore/data/data_utils.py:41-60)."""
    return _collate_padded(values, pad_idx, left_pad, pad_to_length,
                           pad_to_multiple, square=True)


def collate_dict(values, dim=0):
    """Stack same-keyed tensors across a list of dicts."""
    if not values:
        return {}
    return {
        key: torch.stack([entry[key] for entry in values], dim=dim)
        for key in values[0].keys()
    }


@contextlib.contextmanager
def numpy_seed(seed, *addl_seeds):
    """Seed numpy's global PRNG inside the block, then restore the previous
    state (reference unicore/data/data_utils.py:83-104). Extra positional
    seeds are hashed in, which is how per-(epoch, index) determinism is
    derived everywhere in the data layer."""
    if seed is None:
        yield
        return
    if addl_seeds:
        seed = int(hash((seed, *addl_seeds)) % 1e8)
    saved = np.random.get_state()
    np.random.seed(seed)
    try:
        yield
    finally:
        np.random.set_state(saved)


def batch_by_size(indices, batch_size=None, required_batch_size_multiple=1):
    """Split ``indices`` into fixed-size batches
    (reference unicore/data/data_utils.py:107-139).

    The stride is ``batch_size`` rounded up to the multiple; only the final
    short batch may be smaller.
    """
    batch_size = 1 if batch_size is None else batch_size
    stride = -(-batch_size // required_batch_size_multiple)
    stride *= required_batch_size_multiple

    if not isinstance(indices, np.ndarray):
        indices = np.fromiter(indices, dtype=np.int64, count=-1)

    count = -(-len(indices) // stride)
    cuts = stride * (1 + np.arange(count - 1))
    batches = np.split(indices, cuts)
    assert len(batches) == count
    # validation or test data size might be smaller than one mini-batch
    assert batch_size <= 0 or len(batches[0]) <= batch_size
    return batches
