"""Single-file key-value dataset: the lmdb-free storage backend.

Role parity with LMDBDataset (reference unicore/data/lmdb_dataset.py):
pickled values behind integer keys, lazily opened per DataLoader worker.
Where the reference assumes the ``lmdb`` package, this format is a plain
append-only file we read with mmap — no external dependency, which matters
on ROCm images that ship without lmdb.

Layout:
    8 bytes   magic  b"UNIKV001"
    8 bytes   little-endian u64 item count N
    N * 16    (offset u64, length u64) table
    payload   pickled blobs back to back
"""

import mmap
import os
import pickle
import struct

from .unicore_dataset import UnicoreDataset

_MAGIC = b"UNIKV001"
_HEAD = struct.Struct("<8sQ")
_SLOT = struct.Struct("<QQ")


class KVWriter:
    """Sequential writer; use as a context manager."""

    def __init__(self, path):
        self.path = path
        self._blobs = []

    def put(self, obj) -> None:
        self._blobs.append(pickle.dumps(obj, protocol=4))

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc, tb):
        if exc_type is None:
            self.flush()

    def flush(self) -> None:
        table_at = _HEAD.size
        payload_at = table_at + _SLOT.size * len(self._blobs)
        with open(self.path, "wb") as f:
            f.write(_HEAD.pack(_MAGIC, len(self._blobs)))
            cursor = payload_at
            for blob in self._blobs:
                f.write(_SLOT.pack(cursor, len(blob)))
                cursor += len(blob)
            for blob in self._blobs:
                f.write(blob)


class KVDataset(UnicoreDataset):
    """Read side: mmap the file, unpickle on demand."""

    def __init__(self, path):
        assert os.path.isfile(path), f"{path} not found"
        self.path = path
        self._mm = None
        with open(path, "rb") as f:
            magic, count = _HEAD.unpack(f.read(_HEAD.size))
            assert magic == _MAGIC, f"{path}: not a UNIKV file"
            self._count = count

    def _ensure_open(self):
        if self._mm is None:
            # per-worker lazy mmap (the object pickles across fork cleanly)
            f = open(self.path, "rb")
            self._mm = mmap.mmap(f.fileno(), 0, access=mmap.ACCESS_READ)

    def __len__(self):
        return self._count

    def __getitem__(self, idx):
        if not 0 <= idx < self._count:
            raise IndexError(idx)
        self._ensure_open()
        slot_at = _HEAD.size + _SLOT.size * idx
        offset, length = _SLOT.unpack(
            self._mm[slot_at: slot_at + _SLOT.size]
        )
        return pickle.loads(self._mm[offset: offset + length])

    def __getstate__(self):
        state = dict(self.__dict__)
        state["_mm"] = None  # re-open in the worker
        return state
