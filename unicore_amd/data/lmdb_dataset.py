"""LMDB-backed dataset (parity: reference unicore/data/lmdb_dataset.py:16-50).

Values are pickled python objects keyed by the database's native byte keys.
The environment handle is NOT opened in __init__ (beyond a one-shot key
scan): DataLoader workers each lazily open their own env on first access,
keeping the dataset object picklable.
"""

import logging
import os
import pickle
from functools import lru_cache

from .unicore_dataset import UnicoreDataset

logger = logging.getLogger(__name__)

_ENV_OPTS = dict(subdir=False, readonly=True, lock=False, readahead=False,
                 meminit=False, max_readers=256)


class LMDBDataset(UnicoreDataset):
    def __init__(self, db_path):
        try:
            import lmdb  # noqa: F401
        except ImportError:
            raise ImportError(
                "LMDBDataset requires the `lmdb` package (pip install lmdb)"
            )
        assert os.path.isfile(db_path), f"{db_path} not found"
        self.db_path = db_path
        self.env = None
        # one-shot key scan with a throwaway env (keys define len + order)
        scan_env = self._open()
        with scan_env.begin() as txn:
            self._keys = list(txn.cursor().iternext(values=False))
        scan_env.close()

    def _open(self):
        import lmdb

        return lmdb.open(self.db_path, **_ENV_OPTS)

    def connect_db(self, lmdb_path, save_to_self=False):
        """Open an env for *lmdb_path* (kept for reference-API parity;
        reference unicore/data/lmdb_dataset.py:27-43)."""
        import lmdb

        handle = lmdb.open(lmdb_path, **_ENV_OPTS)
        if not save_to_self:
            return handle
        self.env = handle

    def __len__(self):
        return len(self._keys)

    @lru_cache(16)
    def __getitem__(self, idx):
        if self.env is None:
            self.env = self._open()  # per-worker lazy open
        blob = self.env.begin().get(self._keys[idx])
        return pickle.loads(blob)
