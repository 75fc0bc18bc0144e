"""Task base class (parity: reference unicore/tasks/unicore_task.py:20-329).

A task owns the dictionaries and datasets, builds the model/loss from the
registries, constructs the resumable batch iterator, and hosts the per-step
forward/backward delegation the trainer calls into.
"""

import logging
import os
from argparse import Namespace
from typing import Any, Callable, Dict

import torch

from unicore_amd.data import UnicoreDataset, data_utils, iterators
from unicore_amd.logging import metrics

logger = logging.getLogger(__name__)

_warned_once = set()


def warnings_once(msg):
    """Log each distinct warning message a single time per process."""
    if msg not in _warned_once:
        _warned_once.add(msg)
        logger.warning(msg)


class StatefulContainer:
    """Lazily-materialized, checkpointable task state
    (reference unicore/tasks/unicore_task.py:20-42). Attributes resolve
    against stored state first, then registered factories."""

    def __init__(self):
        self._state = {}
        self._factories = {}

    def add_factory(self, name, factory: Callable[[], Any]):
        self._factories[name] = factory

    def merge_state_dict(self, state_dict: Dict[str, Any]):
        self._state.update(state_dict)

    @property
    def state_dict(self) -> Dict[str, Any]:
        return self._state

    def __getattr__(self, name):
        if name not in self._state and name in self._factories:
            self._state[name] = self._factories[name]()
        if name in self._state:
            return self._state[name]
        raise AttributeError(f"Task state has no factory for attribute {name}")


class UnicoreTask:
    """Owns datasets + dictionaries; builds models/losses; drives steps."""

    @classmethod
    def add_args(cls, parser):
        """Hook for task-specific CLI arguments."""

    @staticmethod
    def logging_outputs_can_be_summed(loss, is_train) -> bool:
        """Forwarded from the loss: True enables the fast all-reduce stat
        sync across ranks."""
        return loss.logging_outputs_can_be_summed(is_train)

    def __init__(self, args: Namespace, **kwargs):
        self.args = args
        self.datasets = {}
        self.dataset_to_epoch_iter = {}
        self.state = StatefulContainer()

    @classmethod
    def setup_task(cls, args: Namespace, **kwargs):
        """Factory hook (load dictionaries etc.) before construction."""
        return cls(args, **kwargs)

    def has_sharded_data(self, split):
        return os.pathsep in getattr(self.args, "data", "")

    def load_dataset(self, split: str, combine: bool = False, **kwargs):
        """Load one named split (train/valid/test) into self.datasets."""
        raise NotImplementedError("tasks implement load_dataset")

    def dataset(self, split):
        """The loaded dataset for *split* (must be a UnicoreDataset)."""
        if split not in self.datasets:
            raise KeyError("Dataset not loaded: " + split)
        loaded = self.datasets[split]
        if not isinstance(loaded, UnicoreDataset):
            raise TypeError("Datasets are expected to be of type UnicoreDataset")
        return loaded

    def can_reuse_epoch_itr(self, dataset):
        """Datasets whose item sizes are epoch-stable opt in via the
        can_reuse_epoch_itr_across_epochs property."""
        return getattr(dataset, "can_reuse_epoch_itr_across_epochs", False)

    def get_batch_iterator(
        self,
        dataset,
        batch_size=None,
        ignore_invalid_inputs=False,
        required_batch_size_multiple=1,
        seed=1,
        num_shards=1,
        shard_id=0,
        num_workers=0,
        epoch=1,
        data_buffer_size=0,
        disable_iterator_cache=False,
    ):
        """Build (or reuse) the epoch iterator over *dataset*
        (reference unicore/tasks/unicore_task.py:138-225): ordered indices
        under the seed, fixed-size batches, then a sharded resumable
        EpochBatchIterator."""
        reusable = (
            not disable_iterator_cache and self.can_reuse_epoch_itr(dataset)
        )
        if reusable and dataset in self.dataset_to_epoch_iter:
            logger.debug(f"reusing EpochBatchIterator for epoch {epoch}")
            return self.dataset_to_epoch_iter[dataset]

        assert isinstance(dataset, UnicoreDataset)
        dataset.set_epoch(epoch)  # epoch-aware noising starts correctly

        with data_utils.numpy_seed(seed):
            indices = dataset.ordered_indices()
        batches = dataset.batch_by_size(
            indices,
            batch_size=batch_size,
            required_batch_size_multiple=required_batch_size_multiple,
        )

        epoch_iter = iterators.EpochBatchIterator(
            dataset=dataset,
            collate_fn=dataset.collater,
            batch_sampler=batches,
            seed=seed,
            num_shards=num_shards,
            shard_id=shard_id,
            num_workers=num_workers,
            epoch=epoch,
            buffer_size=data_buffer_size,
            # a --cpu run on a CUDA-capable host must not stage batches onto
            # the GPU (the model stays on CPU)
            cuda_prefetch=not getattr(self.args, "cpu", False),
        )
        if reusable:
            self.dataset_to_epoch_iter[dataset] = epoch_iter
        return epoch_iter

    def build_model(self, args: Namespace):
        """Instantiate this task's model via the model registry."""
        from unicore_amd import models

        return models.build_model(args, self)

    def build_loss(self, args: Namespace):
        """Instantiate this task's loss via the loss registry."""
        from unicore_amd import losses

        return losses.build_loss(args, self)

    def train_step(self, sample, model, loss, optimizer, update_num,
                   ignore_grad=False):
        """One micro-batch: forward through *loss*, then backward through
        the optimizer (which applies loss scaling). ``ignore_grad`` zeroes
        the loss so a dummy batch contributes nothing.

        Returns ``(loss, sample_size, logging_output)`` with sample_size
        the gradient denominator.
        """
        model.train()
        model.set_num_updates(update_num)
        with torch.autograd.profiler.record_function("forward"):
            loss_value, sample_size, logging_output = loss(model, sample)
        if ignore_grad:
            loss_value *= 0
        with torch.autograd.profiler.record_function("backward"):
            optimizer.backward(loss_value)
        return loss_value, sample_size, logging_output

    def valid_step(self, sample, model, loss, test=False):
        model.eval()
        with torch.no_grad():
            return loss(model, sample)

    def optimizer_step(self, optimizer, model, update_num):
        optimizer.step()

    def reduce_metrics(self, logging_outputs, loss, split="train"):
        """Fold per-rank logging outputs into the metrics aggregators."""
        if any("bsz" in log for log in logging_outputs):
            bsz = sum(log.get("bsz", 0) for log in logging_outputs)
            metrics.log_scalar("bsz", bsz, priority=190, round=1)
        else:
            warnings_once(
                "bsz not found in Loss logging outputs, cannot log bsz"
            )
        loss.__class__.reduce_metrics(logging_outputs, split)

    def state_dict(self) -> Dict[str, Any]:
        return self.state.state_dict if self.state is not None else {}

    def load_state_dict(self, state_dict: Dict[str, Any]):
        if self.state is not None:
            self.state.merge_state_dict(state_dict)

    def disable_shuffling(self) -> bool:
        return False

    def begin_epoch(self, epoch, model):
        """Hook invoked as each training epoch starts."""

    def begin_valid_epoch(self, epoch, model):
        """Hook invoked as each validation pass starts."""
