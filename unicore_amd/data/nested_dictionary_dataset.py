"""Composite dataset over a nested dict of datasets (parity: reference
unicore/data/nested_dictionary_dataset.py:48-111).

The nested definition is flattened to dotted keys ("net_input.src_tokens");
items and collated batches are produced flat, then re-nested on the way out.
"""

from collections import OrderedDict

import torch
from torch.utils.data.dataloader import default_collate

from .unicore_dataset import UnicoreDataset


def _flatten(tree, prefix=None):
    """Depth-first flatten of nested dict/list structure into dotted keys."""
    flat = OrderedDict()
    if isinstance(tree, dict):
        head = "" if prefix is None else prefix + "."
        for key, sub in tree.items():
            if sub is None:
                continue
            flat.update(_flatten(sub, head + key))
    elif isinstance(tree, list):
        for pos, sub in enumerate(tree):
            flat.update(_flatten(sub, f"{prefix}.[{pos}]"))
    else:
        flat[prefix] = tree
    return flat


def _unflatten(flat):
    """Rebuild the nested structure from dotted keys."""
    tree = OrderedDict()
    for dotted, value in flat.items():
        *path, leaf = dotted.split(".")
        node = tree
        for part in path:
            if part[:1] == "[" and part[-1:] == "]":
                part = int(part[1:-1])
            node = node.setdefault(part, OrderedDict())
        node[leaf] = value
    return tree


class NestedDictionaryDataset(UnicoreDataset):
    def __init__(self, defn):
        super().__init__()
        self.defn = _flatten(defn)

        first = None
        for leaf in self.defn.values():
            if not isinstance(leaf, (UnicoreDataset, torch.utils.data.Dataset)):
                raise ValueError(f"Expected Dataset but found: {leaf.__class__}")
            first = first or leaf
            if len(leaf) > 0:
                assert len(leaf) == len(first), "dataset lengths must match"
        self._len = len(first)

    def __getitem__(self, index):
        return OrderedDict((k, ds[index]) for k, ds in self.defn.items())

    def __len__(self):
        return self._len

    def collater(self, samples):
        """Collate each leaf with its own collater (default_collate where a
        leaf has none), then re-nest."""
        if len(samples) == 0:
            return {}
        batch = OrderedDict()
        for key, ds in self.defn.items():
            column = [s[key] for s in samples]
            try:
                batch[key] = ds.collater(column)
            except NotImplementedError:
                batch[key] = default_collate(column)
        return _unflatten(batch)

    @property
    def supports_prefetch(self):
        return any(ds.supports_prefetch for ds in self.defn.values())

    def prefetch(self, indices):
        for ds in self.defn.values():
            if getattr(ds, "supports_prefetch", False):
                ds.prefetch(indices)

    @property
    def can_reuse_epoch_itr_across_epochs(self):
        return all(
            ds.can_reuse_epoch_itr_across_epochs for ds in self.defn.values()
        )

    def set_epoch(self, epoch):
        super().set_epoch(epoch)
        for ds in self.defn.values():
            ds.set_epoch(epoch)
