"""Task base class (parity: reference unicore/tasks/unicore_task.py:20-329)."""

import logging
import os
from argparse import Namespace
from typing import Any, Callable, Dict

import torch

from unicore_amd.data import UnicoreDataset, data_utils, iterators
from unicore_amd.logging import metrics

logger = logging.getLogger(__name__)


class StatefulContainer(object):
    """Checkpointable task state (reference unicore/tasks/unicore_task.py:20-42)."""

    def __init__(self):
        self._state = dict()
        self._factories = dict()

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


class UnicoreTask(object):
    """
    Tasks store dictionaries and provide helpers for loading/iterating over
    Datasets, initializing the Model/Loss and calculating the loss.
    """

    @classmethod
    def add_args(cls, parser):
        """Add task-specific arguments to the parser."""
        pass

    @staticmethod
    def logging_outputs_can_be_summed(loss, is_train) -> bool:
        """
        Whether the logging outputs returned by `train_step` and `valid_step` can
        be summed across workers prior to calling `aggregate_logging_outputs`.
        Setting this to True will improve distributed training speed.
        """
        return loss.logging_outputs_can_be_summed(is_train)

    def __init__(self, args: Namespace, **kwargs):
        self.args = args
        self.datasets = dict()
        self.dataset_to_epoch_iter = dict()
        self.state = StatefulContainer()

    @classmethod
    def setup_task(cls, args: Namespace, **kwargs):
        """Setup the task (e.g., load dictionaries)."""
        return cls(args, **kwargs)

    def has_sharded_data(self, split):
        return os.pathsep in getattr(self.args, "data", "")

    def load_dataset(
        self,
        split: str,
        combine: bool = False,
        **kwargs,
    ):
        """Load a given dataset split.

        Args:
            split (str): name of the split (e.g., train, valid, test)
        """
        raise NotImplementedError

    def dataset(self, split):
        """Return a loaded dataset split."""
        from unicore_amd.data import UnicoreDataset

        if split not in self.datasets:
            raise KeyError("Dataset not loaded: " + split)
        if not isinstance(self.datasets[split], UnicoreDataset):
            raise TypeError("Datasets are expected to be of type UnicoreDataset")
        return self.datasets[split]

    def can_reuse_epoch_itr(self, dataset):
        # We can reuse the epoch iterator across epochs as long as the dataset
        # hasn't disabled it. We default to ``False`` here, although in practice
        # this will be ``True`` for most datasets that don't use noising.
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
        """
        Get an iterator that yields batches of data from the given dataset.
        (reference unicore/tasks/unicore_task.py:138-225)
        """
        can_reuse_epoch_itr = not disable_iterator_cache and self.can_reuse_epoch_itr(
            dataset
        )
        if can_reuse_epoch_itr and dataset in self.dataset_to_epoch_iter:
            logger.debug("reusing EpochBatchIterator for epoch {}".format(epoch))
            return self.dataset_to_epoch_iter[dataset]

        assert isinstance(dataset, UnicoreDataset)

        # initialize the dataset with the correct starting epoch
        dataset.set_epoch(epoch)

        # get indices ordered by example size
        with data_utils.numpy_seed(seed):
            indices = dataset.ordered_indices()

        # create mini-batches with given size constraints
        batch_sampler = dataset.batch_by_size(
            indices,
            batch_size=batch_size,
            required_batch_size_multiple=required_batch_size_multiple,
        )

        # return a reusable, sharded iterator
        epoch_iter = iterators.EpochBatchIterator(
            dataset=dataset,
            collate_fn=dataset.collater,
            batch_sampler=batch_sampler,
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

        if can_reuse_epoch_itr:
            self.dataset_to_epoch_iter[dataset] = epoch_iter

        return epoch_iter

    def build_model(self, args: Namespace):
        """Build the :class:`~unicore_amd.models.BaseUnicoreModel` instance for
        this task."""
        from unicore_amd import models

        model = models.build_model(args, self)
        return model

    def build_loss(self, args: Namespace):
        """Build the :class:`~unicore_amd.losses.UnicoreLoss` instance for this
        task."""
        from unicore_amd import losses

        loss = losses.build_loss(args, self)
        return loss

    def train_step(
        self, sample, model, loss, optimizer, update_num, ignore_grad=False
    ):
        """
        Do forward and backward, and return the loss as computed by *loss*
        for the given *model* and *sample*.

        Args:
            sample (dict): the mini-batch. The format is defined by the
                :class:`~unicore_amd.data.UnicoreDataset`.
            model (~unicore_amd.models.BaseUnicoreModel): the model
            loss (~unicore_amd.losses.UnicoreLoss): the loss
            optimizer (~unicore_amd.optim.UnicoreOptimizer): the optimizer
            update_num (int): the current update
            ignore_grad (bool): multiply loss by 0 if this is set to True

        Returns:
            tuple:
                - the loss
                - the sample size, which is used as the denominator for the
                  gradient
                - logging outputs to display while training
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
            loss_value, sample_size, logging_output = loss(model, sample)
        return loss_value, sample_size, logging_output

    def optimizer_step(self, optimizer, model, update_num):
        optimizer.step()

    def reduce_metrics(self, logging_outputs, loss, split="train"):
        """Aggregate logging outputs from data parallel training."""
        if not any("bsz" in log for log in logging_outputs):
            warnings_once(
                "bsz not found in Loss logging outputs, cannot log bsz"
            )
        else:
            bsz = sum(log.get("bsz", 0) for log in logging_outputs)
            metrics.log_scalar("bsz", bsz, priority=190, round=1)

        loss.__class__.reduce_metrics(logging_outputs, split)

    def state_dict(self):
        if self.state is not None:
            return self.state.state_dict
        return {}

    def load_state_dict(self, state_dict: Dict[str, Any]):
        if self.state is not None:
            self.state.merge_state_dict(state_dict)

    def disable_shuffling(self) -> bool:
        return False

    def begin_epoch(self, epoch, model):
        """Hook function called before the start of each epoch."""
        pass

    def begin_valid_epoch(self, epoch, model):
        """Hook function called before the start of each validation epoch."""
        pass


_warned = set()


def warnings_once(msg):
    if msg not in _warned:
        _warned.add(msg)
        logger.warning(msg)
