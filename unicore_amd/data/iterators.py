"""Epoch/batch iterators: resumable, sharded, grouped, background-prefetched.

Capability parity with the reference iterator stack (unicore/data/iterators.py
— CountingIterator:28, EpochBatchIterator:151, GroupedIterator:406,
ShardedIterator:438, BackgroundConsumer:471, BufferedIterator:496), written
around three building blocks:

* :class:`CountingIterator` — position/len bookkeeping shared by everything;
* :class:`EpochBatchIterator` — the resumable multi-epoch front end
  (state_dict round-trips through checkpoints, including the proportional
  position remap when the world size changes);
* the prefetch tail — :class:`BufferedIterator` (thread ahead of the loop)
  and the MI355X-specific :class:`CudaPrefetcher`, which lands each pinned
  batch on the GPU via a dedicated HIP copy stream one step early instead of
  paying a synchronous H2D at sample-prep time.
"""

import itertools
import logging
import math
import os
import queue
import threading
import time

import numpy as np
import torch

from . import data_utils

logger = logging.getLogger(__name__)

# queue marker: the producer thread finished the epoch
_DONE = object()


class CountingIterator:
    """Iterator wrapper that tracks how many items were consumed.

    ``n`` is the consumed count; ``total`` bounds the length (and can be
    lowered with :meth:`take`). ``start`` seeds the count without advancing
    the underlying iterable — used when resuming mid-epoch.
    """

    def __init__(self, iterable, start=None, total=None):
        self.iterable = iterable
        self.n = getattr(iterable, "n", 0) if start is None else start
        self.total = self.n + len(iterable) if total is None else total
        self.itr = self._produce()

    def _produce(self):
        for item in self.iterable:
            if self.n >= self.total:
                raise RuntimeError(
                    "Mismatch between actual and expected iterable length. "
                    "This may be caused by resuming from a checkpoint with a "
                    "different batch configuration."
                )
            self.n += 1
            yield item

    def __iter__(self):
        return self.itr

    def __next__(self):
        return next(self.itr)

    def __len__(self) -> int:
        return self.total

    def has_next(self) -> bool:
        """True while elements remain."""
        return self.n < self.total

    def skip(self, num_to_skip):
        """Advance past *num_to_skip* elements and return self."""
        next(itertools.islice(self.itr, num_to_skip, num_to_skip), None)
        return self

    def take(self, n) -> None:
        """Cap the iterator at *n* elements total."""
        self.total = min(self.total, n)
        if hasattr(self.iterable, "take"):
            self.iterable.take(n)  # propagate the cap downward
        else:
            self.iterable = itertools.islice(self.iterable, n)


class EpochBatchIterating:
    """Interface of epoch-capable batch iterators (see EpochBatchIterator)."""

    def _abstract(self):
        raise NotImplementedError("implemented by EpochBatchIterator")

    def __len__(self) -> int:
        self._abstract()

    @property
    def next_epoch_idx(self):
        self._abstract()

    def next_epoch_itr(self, shuffle=True, fix_batches_to_gpus=False):
        self._abstract()

    def end_of_epoch(self) -> bool:
        self._abstract()

    @property
    def iterations_in_epoch(self) -> int:
        self._abstract()

    def state_dict(self) -> dict:
        self._abstract()

    def load_state_dict(self, state: dict) -> None:
        self._abstract()


class EpochBatchIterator(EpochBatchIterating):
    """Resumable multi-epoch iterator over a torch Dataset.

    Differences from a plain DataLoader: reusable across epochs via
    :meth:`next_epoch_itr` (with per-epoch seeded shuffling), serializable
    mid-epoch via state_dict/load_state_dict, and sharded across data-parallel
    ranks (``num_shards``/``shard_id``).

    Parameters mirror the reference contract
    (unicore/data/iterators.py:151-225): *batch_sampler* is either a frozen
    list of index batches or a callable re-evaluated each epoch;
    *buffer_size* (capped at 32) enables the background prefetch thread;
    *cuda_prefetch* opts into the GPU-side copy-stream staging and must be
    False for CPU-only runs even on a CUDA-capable host.
    """

    def __init__(
        self,
        dataset,
        collate_fn,
        batch_sampler,
        seed=1,
        num_shards=1,
        shard_id=0,
        num_workers=0,
        epoch=1,
        buffer_size=0,
        timeout=0,
        disable_shuffling=False,
        cuda_prefetch=True,
    ):
        assert isinstance(dataset, torch.utils.data.Dataset)
        self.dataset = dataset
        self.collate_fn = collate_fn
        self.batch_sampler = batch_sampler
        self._frozen_batches = (
            None if callable(batch_sampler) else tuple(batch_sampler)
        )
        self.seed = seed
        self.num_shards = num_shards
        self.shard_id = shard_id
        self.num_workers = num_workers
        self.buffer_size = min(buffer_size, 32)  # bound host memory use
        self.timeout = timeout
        self.disable_shuffling = disable_shuffling
        self.cuda_prefetch = cuda_prefetch and torch.cuda.is_available()

        self.epoch = max(epoch, 1)  # epochs are 1-based
        self.shuffle = not disable_shuffling
        self._cur_epoch_itr = None
        self._next_epoch_itr = None
        self._supports_prefetch = getattr(dataset, "supports_prefetch", False)

    @property
    def frozen_batches(self):
        if self._frozen_batches is None:
            self._frozen_batches = tuple(
                self.batch_sampler(self.dataset, self.epoch)
            )
        return self._frozen_batches

    @property
    def first_batch(self):
        if len(self.frozen_batches) == 0:
            raise Exception(
                "The dataset is empty. This could indicate "
                "that all elements in the dataset have been skipped. "
                "Try increasing the max number of allowed tokens or using "
                "a larger dataset."
            )
        if getattr(self.dataset, "supports_fetch_outside_dataloader", True):
            rows = [self.dataset[i] for i in self.frozen_batches[0]]
            return self.collate_fn(rows)
        return "DUMMY"

    def __len__(self) -> int:
        return -(-len(self.frozen_batches) // self.num_shards)

    @property
    def n(self):
        return self.iterations_in_epoch

    @property
    def next_epoch_idx(self):
        """Epoch number the next next_epoch_itr() call will serve."""
        if self._next_epoch_itr is not None:
            return self.epoch  # a resume already queued this epoch
        if self._cur_epoch_itr is not None and self.end_of_epoch():
            return self.epoch + 1
        return self.epoch

    def next_epoch_itr(self, shuffle=True, fix_batches_to_gpus=False):
        """Start (or resume) the next epoch and return its iterator.

        *fix_batches_to_gpus* keeps batch->shard assignment stable across
        epochs (requires dataset prefetch support); shuffling then happens
        within the shard.
        """
        if self.disable_shuffling:
            shuffle = False
        self.epoch = self.next_epoch_idx
        if hasattr(self.dataset, "set_epoch"):
            self.dataset.set_epoch(self.epoch)
        if self._next_epoch_itr is not None:
            # a load_state_dict() prepared a mid-epoch iterator
            self._cur_epoch_itr = self._next_epoch_itr
            self._next_epoch_itr = None
        else:
            if callable(self.batch_sampler):
                self._frozen_batches = None  # regenerate for the new epoch
            self._cur_epoch_itr = self._build_epoch_itr(
                self.epoch, shuffle, fix_batches_to_gpus=fix_batches_to_gpus
            )
        self.shuffle = shuffle
        return self._cur_epoch_itr

    def end_of_epoch(self) -> bool:
        return not self._cur_epoch_itr.has_next()

    @property
    def iterations_in_epoch(self) -> int:
        for itr in (self._cur_epoch_itr, self._next_epoch_itr):
            if itr is not None:
                return itr.n
        return 0

    def state_dict(self) -> dict:
        if self.end_of_epoch():
            # finished epochs serialize as the start of the following one
            epoch, offset = self.epoch + 1, 0
        else:
            epoch, offset = self.epoch, self.iterations_in_epoch
        return {
            "version": 2,
            "epoch": epoch,
            "iterations_in_epoch": offset,
            "shuffle": self.shuffle,
            "len": len(self),
        }

    def load_state_dict(self, state: dict) -> None:
        self.epoch = state["epoch"]
        offset = state.get("iterations_in_epoch", 0)
        version = state.get("version", 1)
        if offset <= 0:
            self._next_epoch_itr = None
            return
        saved_len = state.get("len", len(self))
        if saved_len != len(self):
            # world size or batching changed across the restart: map the
            # position proportionally into the new batch count
            # (reference unicore/data/iterators.py:326-350)
            remapped = int(math.floor(len(self) * offset / saved_len))
            logger.info(
                f"Iterator length changed from {saved_len} to {len(self)}; "
                f"proportionally remapping resume position to {remapped}"
            )
            offset = remapped
        self._next_epoch_itr = self._build_epoch_itr(
            self.epoch, shuffle=state.get("shuffle", True), offset=offset
        )
        if self._next_epoch_itr is None:
            if version == 1:
                # v1 checkpoints treated a past-the-end offset as a completed
                # epoch
                self.epoch += 1
            else:
                raise RuntimeError(
                    "Cannot resume training due to dataloader mismatch. You "
                    "can relaunch training with `--reset-dataloader` and it "
                    "should work."
                )

    # -- epoch iterator construction --------------------------------------

    def _shuffled(self, batches, seed):
        batches = list(batches)
        with data_utils.numpy_seed(seed):
            np.random.shuffle(batches)
        return batches

    def _build_epoch_itr(self, epoch, shuffle, fix_batches_to_gpus=False,
                         offset=0):
        if self._supports_prefetch:
            # prefetch-capable datasets want the index list before loading;
            # with fix_batches_to_gpus the shard assignment must not depend
            # on the epoch, so the shuffle happens after sharding with a
            # shard-local seed
            batches = self.frozen_batches
            if shuffle and not fix_batches_to_gpus:
                batches = self._shuffled(batches, self.seed + epoch)
            batches = list(
                ShardedIterator(batches, self.num_shards, self.shard_id,
                                fill_value=[])
            )
            self.dataset.prefetch([i for b in batches for i in b])
            if shuffle and fix_batches_to_gpus:
                batches = self._shuffled(
                    batches, self.seed + epoch + self.shard_id
                )
        else:
            batches = (
                self._shuffled(self.frozen_batches, self.seed + epoch)
                if shuffle
                else self.frozen_batches
            )
            batches = list(
                ShardedIterator(batches, self.num_shards, self.shard_id,
                                fill_value=[])
            )

        if offset > 0 and offset >= len(batches):
            return None  # nothing left of this epoch

        if self.num_workers > 0:
            os.environ["PYTHONWARNINGS"] = "ignore:semaphore_tracker:UserWarning"

        loader = torch.utils.data.DataLoader(
            self.dataset,
            collate_fn=self.collate_fn,
            batch_sampler=batches[offset:],
            num_workers=self.num_workers,
            timeout=self.timeout,
            pin_memory=torch.cuda.is_available(),
        )
        if self.buffer_size > 0:
            loader = BufferedIterator(self.buffer_size, loader)
        if self.cuda_prefetch:
            # inside the CountingIterator so the resume position counts
            # consumed batches, not prefetched ones
            loader = CudaPrefetcher(loader)
        return CountingIterator(loader, start=offset)


def _chunks(itr, size):
    """Yield lists of up to *size* consecutive items."""
    bucket = []
    for item in itr:
        bucket.append(item)
        if len(bucket) == size:
            yield bucket
            bucket = []
    if bucket:
        yield bucket


class GroupedIterator(CountingIterator):
    """Chunks an iterator into lists of *chunk_size* items (the grad-accum
    micro-batch grouping)."""

    def __init__(self, iterable, chunk_size):
        super().__init__(
            _chunks(iterable, chunk_size),
            start=-(-getattr(iterable, "n", 0) // chunk_size),
            total=-(-len(iterable) // chunk_size),
        )
        self.chunk_size = chunk_size


class ShardedIterator(CountingIterator):
    """Round-robin shard of an iterable, padded with *fill_value* so every
    shard sees the same number of batches."""

    def __init__(self, iterable, num_shards, shard_id, fill_value=None):
        if not 0 <= shard_id < num_shards:
            raise ValueError("shard_id must be between 0 and num_shards")
        shard_len = -(-len(iterable) // num_shards)
        strided = itertools.islice(
            iterable, shard_id, len(iterable), num_shards
        )
        padded = (
            pair[1]
            for pair in itertools.zip_longest(
                range(shard_len), strided, fillvalue=fill_value
            )
        )
        super().__init__(
            padded,
            start=-(-getattr(iterable, "n", 0) // num_shards),
            total=shard_len,
        )


class BufferedIterator:
    """Producer-thread lookahead: a daemon thread fills a bounded queue up to
    *size* batches ahead of the consumer, with a rate-limited hint when the
    buffer keeps running dry (loader-bound training)."""

    # warn no earlier than 5 min in, and at most every 15 min
    _WARMUP_S = 5 * 60
    _WARN_EVERY_S = 15 * 60

    def __init__(self, size, iterable):
        self._queue = queue.Queue(size)
        self._iterable = iterable
        self._producer = None
        self.start_time = time.time()
        self.warning_time = None
        self.total = len(iterable)

    def _pump(self, source, limit):
        """Producer thread body: forward items, then signal completion.
        Exceptions travel through the queue to the consumer."""
        try:
            sent = 0
            for item in source:
                self._queue.put(item)
                sent += 1
                if limit is not None and sent >= limit:
                    break
            self._queue.put(_DONE)
        except Exception as exc:  # noqa: BLE001 - must cross the thread
            self._queue.put(exc)

    def _start_producer(self):
        self._producer = threading.Thread(
            target=self._pump, args=(self._iterable, self.total), daemon=True
        )
        self._producer.start()

    def __iter__(self):
        return self

    def __len__(self) -> int:
        return self.total

    def take(self, n) -> None:
        self.total = min(self.total, n)
        if hasattr(self._iterable, "take"):
            self._iterable.take(n)
        else:
            self._iterable = itertools.islice(self._iterable, n)

    def _maybe_warn_starved(self):
        if self._queue.qsize() >= min(2, max(1, self._queue.maxsize // 2)):
            return
        now = time.time()
        if now - self.start_time <= self._WARMUP_S:
            return
        if (self.warning_time is None
                or now - self.warning_time > self._WARN_EVERY_S):
            logger.debug(
                "Data loading buffer is empty or nearly empty. This may "
                "indicate a data loading bottleneck, and increasing the "
                "number of workers (--num-workers) may help."
            )
            self.warning_time = now

    def __next__(self):
        if self._producer is None:
            self._start_producer()
        self._maybe_warn_starved()
        item = self._queue.get(True)
        if isinstance(item, Exception):
            raise item
        if item is _DONE:
            raise StopIteration()
        return item


class CudaPrefetcher:
    """Overlap H2D copies with compute: move each (pinned) batch to the GPU on
    a dedicated copy stream one batch ahead, and make the consumer's compute
    stream wait on the copy event instead of blocking the host.

    MI355X-native analog of the reference's move_to_cuda(non_blocking=True)
    at sample-prep time (reference unicore/utils.py:64-72 +
    unicore/trainer.py:927-928), but with an explicit side HIP stream so the
    copy engine runs concurrently with the previous step's kernels.
    """

    def __init__(self, iterable, device=None):
        self.iterable = iterable
        self.device = device if device is not None else torch.cuda.current_device()
        self.copy_stream = torch.cuda.Stream(self.device)
        self._next = None
        self._next_event = None
        self._itr = None

    def __len__(self) -> int:
        return len(self.iterable)

    def _preload(self):
        try:
            batch = next(self._itr)
        except StopIteration:
            self._next = None
            return
        from ..utils import apply_to_sample

        with torch.cuda.stream(self.copy_stream):
            self._next = apply_to_sample(
                lambda t: t.to(device=self.device, non_blocking=True), batch
            )
            self._next_event = torch.cuda.Event()
            self._next_event.record(self.copy_stream)

    def __iter__(self):
        from ..utils import apply_to_sample

        def _claim(t):
            # the tensors were allocated on the copy stream; mark them as
            # used by the consumer stream so the caching allocator does not
            # hand their memory to a LATER prefetch copy while this step's
            # kernels are still reading it (classic cross-stream free race)
            t.record_stream(torch.cuda.current_stream())
            return t

        self._itr = iter(self.iterable)
        self._preload()
        while self._next is not None:
            batch = self._next
            event = self._next_event
            self._preload()
            # consumer stream waits for the copy to land; tensors stay alive
            # via the yielded reference
            event.wait(torch.cuda.current_stream())
            batch = apply_to_sample(_claim, batch)
            yield batch
