"""LMDB-backed dataset (parity: reference unicore/data/lmdb_dataset.py:16-50).

Values are pickled python objects; the env is opened lazily per worker so the
dataset object itself stays picklable for DataLoader workers.
"""

import logging
import os
import pickle

from functools import lru_cache

from .unicore_dataset import UnicoreDataset

logger = logging.getLogger(__name__)


class LMDBDataset(UnicoreDataset):
    def __init__(self, db_path):
        try:
            import lmdb  # noqa: F401
        except ImportError:
            raise ImportError(
                "LMDBDataset requires the `lmdb` package (pip install lmdb)"
            )
        self.db_path = db_path
        assert os.path.isfile(self.db_path), "{} not found".format(self.db_path)
        env = self.connect_db(self.db_path)
        with env.begin() as txn:
            self._keys = list(txn.cursor().iternext(values=False))
        env.close()
        self.env = None

    def connect_db(self, lmdb_path, save_to_self=False):
        import lmdb

        env = lmdb.open(
            lmdb_path,
            subdir=False,
            readonly=True,
            lock=False,
            readahead=False,
            meminit=False,
            max_readers=256,
        )
        if not save_to_self:
            return env
        else:
            self.env = env

    def __len__(self):
        return len(self._keys)

    @lru_cache(maxsize=16)
    def __getitem__(self, idx):
        if self.env is None:
            self.connect_db(self.db_path, save_to_self=True)
        datapoint_pickled = self.env.begin().get(self._keys[idx])
        data = pickle.loads(datapoint_pickled)
        return data
