"""Trainer: owns model/loss/optimizer/lr-scheduler/EMA and runs the hot loop.

Parity: reference unicore/trainer.py — __init__:40, train_step:571-802,
valid_step:804-848, _build_optimizer:220-256, state_dict:258-284,
save/load_checkpoint:286-482, get_train_iterator:484, aggregate logging:
967-1009, grad-norm consistency check:1051-1084.

MI355X specifics: the DDP engine is our FlatDDP (bucketed all-reduce on a
side HIP stream overlapped with backward, buckets aliased onto the fp16/bf16
optimizer's flat grad buffers), gradient H2D staging uses pinned
non_blocking copies, and rocTX-visible phase names are kept identical to the
reference's record_function annotations so rocprof traces line up.

Update anatomy (one train_step):
  micro-batch loop (no_sync on all but the last) -> cross-rank stat sync ->
  reduce-grads -> multiply-grads (world/sample_size, undoes loss scale) ->
  clip-grads -> cross-rank grad-norm consistency check -> optimizer (rank-
  identical seed for SR) -> EMA. Overflow skips the update and rescales;
  NaN/Inf triggers a forensic NanDetector re-run.
"""

import contextlib
import logging
import os
import sys
import time
from itertools import chain
from typing import Any, Dict, List

import torch

from unicore_amd import checkpoint_utils, models, optim, utils
from unicore_amd.distributed import utils as distributed_utils
from unicore_amd.ema import ExponentialMovingAverageModel
from unicore_amd.logging import meters, metrics
from unicore_amd.nan_detector import NanDetector
from unicore_amd.optim import lr_scheduler

logger = logging.getLogger(__name__)


def _looks_like_oom(exc):
    return "out of memory" in str(exc)


def _zeroed_stat(stat):
    """Zero a (tensor or plain) accumulated statistic in place-ish."""
    if torch.is_tensor(stat):
        stat.zero_()
        return stat
    return stat * 0.0


def _find_tied_params(module, memo=None, prefix=""):
    """Parameter objects reachable under more than one dotted path
    (reference unicore/trainer.py:1127-1145). Walks raw _parameters because
    named_parameters() deduplicates exactly the sharing we want to see."""
    root_call = memo is None
    if root_call:
        memo = {}
    for name, param in module._parameters.items():
        dotted = f"{prefix}.{name}" if prefix else name
        memo.setdefault(param, []).append(dotted)
    for name, child in module._modules.items():
        if child is not None:
            _find_tied_params(
                child, memo, f"{prefix}.{name}" if prefix else name
            )
    if root_call:
        return [paths for paths in memo.values() if len(paths) > 1]


def _resolve_attr(module, dotted):
    for part in dotted.split("."):
        module = getattr(module, part)
    return module


def _assign_attr(module, dotted, value):
    *head, leaf = dotted.split(".")
    for part in head:
        module = getattr(module, part)
    setattr(module, leaf, value)


class Trainer:
    """Synchronous data-parallel trainer: every rank holds a full replica;
    gradients are combined across ranks before each update through our
    FlatDDP engine (RCCL over xGMI on an MI355X node)."""

    def __init__(self, args, task, model, loss):
        self.args = args
        self.task = task

        # record tied parameters BEFORE any cast/move breaks the aliasing
        tied = _find_tied_params(model)

        self.cuda = torch.cuda.is_available() and not args.cpu
        self.device = torch.device("cuda" if self.cuda else "cpu")

        # params must reach their final dtype/device BEFORE the optimizer
        # flattens them
        self._model, self._loss = model, loss
        if args.fp16:
            self._model, self._loss = self._model.half(), self._loss.half()
        elif args.bf16:
            self._model = self._model.bfloat16()
            self._loss = self._loss.bfloat16()
        if self.cuda:
            self._model = self._model.to(self.device)
            self._loss = self._loss.to(self.device)

        # re-tie anything .to()/cast may have un-aliased
        for paths in tied:
            canonical = _resolve_attr(self._model, paths[0])
            for other in paths[1:]:
                logger.info(f"detected shared parameter: {paths[0]} <- {other}")
                _assign_attr(self._model, other, canonical)

        self._dummy_batch = None  # filled from the first train iterator
        self._lr_scheduler = None
        self._total_train_steps = None
        self._num_updates = 0
        self._optim_history = None
        self._optimizer = None
        self._warn_once = set()
        self._wrapped_loss = None
        self._wrapped_model = None

        self._criterion_has_params = any(
            p.requires_grad for p in self._loss.parameters()
        )

        self._grad_norm_buf = None
        if self.cuda and self.data_parallel_world_size > 1:
            self._grad_norm_buf = torch.zeros(
                self.data_parallel_world_size, device=self.device
            )

        self.ema = self._build_ema()
        self._snapshot_cuda_env()

        metrics.log_start_time("wall", priority=790, round=2)
        self._start_time = time.time()
        self._previous_training_time = 0
        self._cumulative_training_time = None

    def _build_ema(self):
        """EMA shadow on rank 0 only, unless validation swaps it in on every
        rank (reference unicore/trainer.py:114-130)."""
        args = self.args
        if args.validate_with_ema:
            assert args.ema_decay > 0, "valid with ema must with ema_decay > 0"
        wants_ema = args.ema_decay > 0 and (
            self.data_parallel_rank == 0 or args.validate_with_ema
        )
        if not wants_ema:
            return None
        return ExponentialMovingAverageModel(
            args, self._model, args.ema_decay,
            is_flattened=(args.fp16 or args.bf16),
        )

    def _snapshot_cuda_env(self):
        if not self.cuda:
            self.cuda_env = None
            self.cuda_env_arr = []
            return
        self.cuda_env = utils.CudaEnvironment()
        self.cuda_env_arr = [self.cuda_env]
        if self.data_parallel_world_size > 1:
            self.cuda_env_arr = distributed_utils.all_gather_list(
                self.cuda_env, group=distributed_utils.get_global_group()
            )
        if self.data_parallel_rank == 0:
            utils.CudaEnvironment.pretty_print_cuda_env_list(self.cuda_env_arr)

    def reinitialize(self):
        """Drop derived objects (optimizer/scheduler/wrappers) so they are
        rebuilt lazily, e.g. after swapping model params."""
        self._lr_scheduler = None
        self._optimizer = None
        self._wrapped_loss = None
        self._wrapped_model = None

    # -- topology ----------------------------------------------------------

    @property
    def data_parallel_world_size(self):
        if self.args.distributed_world_size <= 1:
            return 1
        return distributed_utils.get_data_parallel_world_size()

    @property
    def data_parallel_process_group(self):
        return distributed_utils.get_data_parallel_group()

    @property
    def data_parallel_rank(self):
        if self.args.distributed_world_size <= 1:
            return 0
        return distributed_utils.get_data_parallel_rank()

    @property
    def is_data_parallel_master(self):
        return self.data_parallel_rank == 0

    @property
    def use_distributed_wrapper(self) -> bool:
        return self.data_parallel_world_size > 1

    @property
    def should_save_checkpoint_on_current_rank(self) -> bool:
        return self.is_data_parallel_master

    # -- lazily wrapped model/loss/optimizer -------------------------------

    @property
    def loss(self):
        if self._wrapped_loss is None:
            wrap = (
                utils.has_parameters(self._loss)
                and self.use_distributed_wrapper
            )
            self._wrapped_loss = (
                models.DistributedUnicoreModel(
                    self.args, self._loss,
                    process_group=self.data_parallel_process_group,
                    device=self.device,
                )
                if wrap
                else self._loss
            )
        return self._wrapped_loss

    @property
    def model(self):
        if self._wrapped_model is None:
            self._wrapped_model = (
                models.DistributedUnicoreModel(
                    self.args, self._model,
                    process_group=self.data_parallel_process_group,
                    device=self.device,
                )
                if self.use_distributed_wrapper
                else self._model
            )
        return self._wrapped_model

    @property
    def optimizer(self):
        if self._optimizer is None:
            self._build_optimizer()
        return self._optimizer

    @property
    def lr_scheduler(self):
        if self._lr_scheduler is None:
            self._build_optimizer()  # also creates the scheduler
        return self._lr_scheduler

    def _check_mode_constraints(self):
        legacy = ("no_c10d", "legacy_ddp")
        if self.args.per_sample_clip_norm > 0.0:
            assert self.args.ddp_backend in legacy, \
                "--per-sample-clip-norm requires --ddp-backend no_c10d"
            assert self.args.fp16 or self.args.bf16, \
                "--per-sample-clip-norm requires fp16/bf16 training"
        if self.args.allreduce_fp32_grad:
            assert self.args.ddp_backend in legacy, \
                "--allreduce-fp32-grad requires --ddp-backend no_c10d"
            assert self.args.fp16 or self.args.bf16, \
                "--allreduce-fp32-grad requires fp16/bf16 training"

    def _build_optimizer(self):
        trainable = [
            (name, p)
            for name, p in chain(
                self._model.named_parameters(), self._loss.named_parameters()
            )
            if p.requires_grad
        ]
        self._check_mode_constraints()

        low_precision = self.args.fp16 or self.args.bf16
        if not low_precision:
            if self.cuda and torch.cuda.get_device_capability(0)[0] >= 7:
                logger.info("NOTE: your device may support faster training"
                            " with --fp16")
            self._optimizer = optim.build_optimizer(self.args, trainable)
        if low_precision:
            if self.cuda and torch.cuda.get_device_capability(0)[0] < 7:
                logger.info("NOTE: your device does NOT support faster"
                            " training with --fp16, please switch to FP32"
                            " which is likely to be faster")
            self._optimizer = optim.FP16Optimizer.build_optimizer(
                self.args, trainable
            )
            lazy_ok = (
                self.data_parallel_world_size <= 1
                and self.args.per_sample_clip_norm <= 0.0
                and hasattr(self._optimizer, "enable_lazy_grad_collection")
            )
            if lazy_ok:
                # no DDP engine hooks the flat grads: let autograd assign
                # gradients and batch-copy them at sync time instead of
                # one accumulate kernel per parameter per backward
                self._optimizer.enable_lazy_grad_collection()

        # (re)build the DDP wrappers AFTER the optimizer so FlatDDP can
        # alias its buckets onto the optimizer's flat grad buffers; stale
        # wrappers must drop their autograd hooks first
        for wrapped in (self._wrapped_model, self._wrapped_loss):
            inner = getattr(wrapped, "module", None)
            if inner is not None and hasattr(inner, "detach_hooks"):
                inner.detach_hooks()
        self._wrapped_model = None
        self._wrapped_loss = None
        _ = self.model
        _ = self.loss

        # scheduler immediately after the optimizer so the initial LR lands
        self._lr_scheduler = lr_scheduler.build_lr_scheduler(
            self.args, self.optimizer, self._total_train_steps
        )
        self._lr_scheduler.step_update(0)

    def init_total_train_steps(self, epoch_itr):
        """Total updates this run will perform — warmup_ratio schedulers
        need it up front (reference unicore/trainer.py:518-524)."""
        if self.args.max_epoch <= 0:
            self._total_train_steps = self.args.max_update
            return
        per_epoch = (len(epoch_itr) + 1) // max(1, self.args.update_freq[0])
        self._total_train_steps = per_epoch * self.args.max_epoch

    # -- checkpoint state ---------------------------------------------------

    def state_dict(self):
        """Assemble the full checkpoint payload."""
        history = (self._optim_history or []) + [
            {
                "loss_name": type(self.get_loss()).__name__,
                "optimizer_name": type(self.optimizer).__name__,
                "lr_scheduler_state": self.lr_scheduler.state_dict(),
                "num_updates": self.get_num_updates(),
            }
        ]
        payload = {
            "args": self.args,
            "model": self._model.state_dict(),
            "loss": (
                self._loss.state_dict()
                if utils.has_parameters(self._loss)
                else None
            ),
            "optimizer_history": history,
            "task_state": self.task.state_dict() if self.task is not None else {},
            "extra_state": {
                "metrics": metrics.state_dict(),
                "previous_training_time": self.cumulative_training_time(),
            },
        }
        if not self.args.no_save_optimizer_state:
            payload["last_optimizer_state"] = self.optimizer.state_dict()
        if self.ema is not None:
            payload["ema"] = self.ema.state_dict()
        return payload

    def save_checkpoint(self, filename, extra_state):
        """Serialize all training state into *filename*."""
        logger.info(f"Saving checkpoint to {filename}")
        # every rank runs state_dict() (it may communicate); rank 0 writes
        payload = utils.move_to_cpu(self.state_dict())
        payload["extra_state"].update(extra_state)
        if self.should_save_checkpoint_on_current_rank:
            checkpoint_utils.torch_persistent_save(payload, filename)
        logger.info(f"Finished saving checkpoint to {filename}")

    def _fetch_checkpoint_state(self, filename):
        """Rank 0 reads the file; everyone else receives it by broadcast.
        Returns None when the file does not exist anywhere."""
        multi_rank = self.data_parallel_world_size > 1
        exists = os.path.isfile(filename) if self.data_parallel_rank == 0 else None
        if multi_rank:
            exists = distributed_utils.broadcast_object(
                exists, src_rank=0,
                group=self.data_parallel_process_group,
                dist_device=self.device,
            )
        if not exists:
            return None
        state = None
        if self.data_parallel_rank == 0:
            state = checkpoint_utils.load_checkpoint_to_cpu(filename)
        if multi_rank:
            logger.info("Broadcast checkpoint to all ranks")
            state = distributed_utils.broadcast_object(
                state, src_rank=0,
                group=self.data_parallel_process_group,
                dist_device=self.device,
            )
        return state

    def _restore_weights(self, state, filename):
        """Load model (+loss) weights from the checkpoint state; returns
        (ema_state, ema_loaded)."""
        ema_state = state.get("ema", None)
        ema_loaded = False
        try:
            if self.args.load_from_ema:
                logger.info("loading ema state to model")
                assert ema_state is not None, "no EMA state in checkpoint"
                self._model.load_state_dict(
                    ema_state["params"], strict=False, model_args=self.args
                )
                ema_loaded = True
            if not ema_loaded:
                report = self._model.load_state_dict(
                    state["model"], strict=False, model_args=self.args
                )
                for field in ("missing_keys", "unexpected_keys"):
                    keys = getattr(report, field, None) if report is not None else None
                    if keys:
                        logger.warning(
                            f"Error in loading model state, {field} {keys}"
                        )
            del state["model"]  # free before the optimizer state loads

            if utils.has_parameters(self.get_loss()) and state.get("loss"):
                self.get_loss().load_state_dict(state["loss"], strict=True)
            state.pop("loss", None)
        except Exception:
            raise Exception(
                f"Cannot load model parameters from checkpoint {filename}; "
                "please ensure that the architectures match."
            )
        return ema_state, ema_loaded

    def _restore_ema(self, ema_state, ema_loaded):
        if ema_state is not None and self.ema is not None \
                and not self.args.load_from_ema:
            logger.info("loading ema state from checkpoint")
            self.ema.load_state_dict(ema_state)
        elif self.ema is not None and not ema_loaded:
            logger.info(
                "Cannot find EMA state in checkpoint, load model weight to "
                "ema directly"
            )
            self.ema = ExponentialMovingAverageModel(
                self.args, self._model, decay=self.ema.decay,
                is_flattened=(self.args.fp16 or self.args.bf16),
            )

    def load_checkpoint(self, filename, reset_optimizer=False,
                        reset_lr_scheduler=False, optimizer_overrides=None,
                        reset_meters=False, reset_dataloader=False,
                        **passthrough_args):
        """Restore all training state: rank 0 reads + broadcasts, weights
        load everywhere, then the iterator/optimizer/meters are rebuilt.
        Returns (extra_state, epoch_itr)."""
        logger.info(f"Preparing to load checkpoint {filename}")

        extra_state, self._optim_history, last_optim_state = None, [], None
        ema_loaded = False
        state = self._fetch_checkpoint_state(filename)
        if state is not None:
            ema_state, ema_loaded = self._restore_weights(state, filename)
            extra_state = state.get("extra_state", None)
            self._optim_history = state.get("optimizer_history", None)
            last_optim_state = state.get("last_optimizer_state", None)
            self._restore_ema(ema_state, ema_loaded)

        loaded_train_itr = False
        epoch = 1
        if extra_state is not None:
            itr_state = extra_state.get("train_iterator", None)
            if itr_state is not None:
                epoch = itr_state.get("epoch", 1)
                if "previous_training_time" in extra_state:
                    self._previous_training_time = \
                        extra_state["previous_training_time"]
                    self._start_time = time.time()
                if (itr_state.get("version", 1) >= 2
                        and itr_state["iterations_in_epoch"] == 0):
                    reset_meters = True  # clean epoch boundary

            if "metrics" in extra_state and not reset_meters:
                metrics.load_state_dict(extra_state["metrics"])
                # TimeMeters reference a dead process's clock; restart them
                for meter in metrics.get_meters("default"):
                    if isinstance(meter, meters.TimeMeter):
                        meter.reset()

            if itr_state is not None and not reset_dataloader:
                epoch_itr = self.get_train_iterator(
                    epoch=itr_state["epoch"], load_dataset=True,
                    **passthrough_args,
                )
                epoch_itr.load_state_dict(itr_state)
                loaded_train_itr = True

        if not loaded_train_itr:
            epoch_itr = self.get_train_iterator(
                epoch=1, load_dataset=True, **passthrough_args
            )

        self.init_total_train_steps(epoch_itr)

        if last_optim_state is not None and not reset_optimizer:
            # model weights just changed: rebuild before loading opt state
            self._build_optimizer()

            previous = self._optim_history[-1]
            assert previous["loss_name"] == type(self.get_loss()).__name__, (
                "Loss does not match; please reset the optimizer "
                f"(--reset-optimizer). {previous['loss_name']} vs "
                f"{type(self.get_loss()).__name__}"
            )
            assert previous["optimizer_name"] == type(self.optimizer).__name__, (
                "Optimizer does not match; please reset the optimizer "
                f"(--reset-optimizer). {previous['optimizer_name']} vs "
                f"{type(self.optimizer).__name__}"
            )
            if not reset_lr_scheduler:
                self.lr_scheduler.load_state_dict(
                    previous["lr_scheduler_state"]
                )
            self.optimizer.load_state_dict(
                last_optim_state, optimizer_overrides
            )
            self.set_num_updates(previous["num_updates"])

        if extra_state is not None and loaded_train_itr:
            logger.info(
                f"Loaded checkpoint {filename} "
                f"(epoch {epoch} @ {self.get_num_updates()} updates)"
            )
        elif ema_loaded:
            logger.info(f"Loaded ema state from checkpoint {filename}")
        elif extra_state is None:
            logger.info(f"No existing checkpoint found {filename}")

        self.lr_step(epoch_itr.epoch)
        return extra_state, epoch_itr

    # -- iterators ----------------------------------------------------------

    def get_train_iterator(self, epoch, combine=True, load_dataset=True,
                           data_selector=None, shard_batch_itr=True,
                           disable_iterator_cache=False):
        """EpochBatchIterator over the training split for *epoch*."""
        if load_dataset:
            logger.info(f"loading train data for epoch {epoch}")
            self.task.load_dataset(
                self.args.train_subset, epoch=epoch, combine=combine,
                data_selector=data_selector,
            )
        batch_iterator = self.task.get_batch_iterator(
            dataset=self.task.dataset(self.args.train_subset),
            batch_size=self.args.batch_size, ignore_invalid_inputs=True,
            required_batch_size_multiple=(
                self.args.required_batch_size_multiple
            ),
            seed=self.args.seed,
            num_shards=self.data_parallel_world_size if shard_batch_itr else 1,
            shard_id=self.data_parallel_rank if shard_batch_itr else 0,
            num_workers=self.args.num_workers, epoch=epoch,
            data_buffer_size=self.args.data_buffer_size,
            disable_iterator_cache=disable_iterator_cache)
        self._iter_per_epoch = len(batch_iterator)
        self.reset_dummy_batch(batch_iterator.first_batch)
        return batch_iterator

    def get_valid_iterator(self, subset, disable_iterator_cache=False):
        """EpochBatchIterator over one validation subset."""
        batch_iterator = self.task.get_batch_iterator(
            dataset=self.task.dataset(subset),
            batch_size=self.args.batch_size_valid,
            ignore_invalid_inputs=(
                self.args.skip_invalid_size_inputs_valid_test
            ),
            required_batch_size_multiple=(
                self.args.required_batch_size_multiple
            ),
            seed=self.args.seed,
            num_shards=self.data_parallel_world_size,
            shard_id=self.data_parallel_rank,
            num_workers=self.args.num_workers, epoch=1,
            data_buffer_size=self.args.data_buffer_size,
            disable_iterator_cache=disable_iterator_cache)
        self.reset_dummy_batch(batch_iterator.first_batch)
        return batch_iterator

    def begin_epoch(self, epoch):
        logger.info(f"begin training epoch {epoch}")
        self.lr_step_begin_epoch(epoch)
        self.task.begin_epoch(epoch, self.get_model())

    def begin_valid_epoch(self, epoch):
        self.task.begin_valid_epoch(epoch, self.get_model())

    def reset_dummy_batch(self, batch):
        self._dummy_batch = batch

    # -- the hot loop --------------------------------------------------------

    @metrics.aggregate("train")
    def train_step(self, samples, raise_oom=False):
        """Forward + backward over the micro-batch group, then one
        parameter update. Returns the logging output, or None when the
        update was skipped (OOM / overflow)."""
        self._set_seed()
        self.model.train()
        self.loss.train()
        self.zero_grad()

        metrics.log_start_time("train_wall", priority=800, round=2)

        logging_outputs, sample_size, ooms = [], 0, 0
        for i, sample in enumerate(samples):  # grad-accumulation loop
            sample, is_filler = self._prepare_sample(sample)

            def sync_policy():
                # accumulate locally; only the last micro-batch's backward
                # carries the all-reduce
                last = i == len(samples) - 1
                if (self.data_parallel_world_size > 1
                        and hasattr(self.model, "no_sync") and not last):
                    return self.model.no_sync()
                return contextlib.ExitStack()  # no-op context

            try:
                with sync_policy():
                    # per-rank dropout streams, keyed by update + micro-batch
                    with utils.torch_seed(
                        self.args.seed, self.get_num_updates(), i,
                        self.data_parallel_rank,
                    ):
                        loss, micro_size, logging_output = \
                            self.task.train_step(
                                sample=sample, model=self.model,
                                loss=self.loss, optimizer=self.optimizer,
                                update_num=self.get_num_updates(),
                                ignore_grad=is_filler,
                            )
                        psc = self.args.per_sample_clip_norm
                        if psc > 0:
                            self.optimizer.per_sample_clip_grad_norm(psc)
                        del loss

                logging_outputs.append(logging_output)
                sample_size += micro_size

                # flushing after step 0 lowers later OOM risk
                if self.cuda and self.get_num_updates() == 0:
                    torch.cuda.empty_cache()
            except RuntimeError as exc:
                if not _looks_like_oom(exc):
                    raise
                self._log_oom(exc)
                if raise_oom:
                    raise
                logger.warning("attempting to recover from OOM in"
                               " forward/backward pass")
                ooms += 1
                self.zero_grad()
                if self.cuda:
                    torch.cuda.empty_cache()
                if self.args.distributed_world_size <= 1:
                    return None

        if is_filler:
            # dummy batches pad out short epochs; they must not weigh in
            sample_size = _zeroed_stat(sample_size)
        sample_size = (
            sample_size.float()
            if torch.is_tensor(sample_size)
            else float(sample_size)
        )

        if self._sync_stats():
            train_time = self._local_cumulative_training_time()
            logging_outputs, (
                sample_size, ooms, total_train_time,
            ) = self._aggregate_logging_outputs(
                logging_outputs, sample_size, ooms, train_time,
                ignore=is_filler,
            )
            self._cumulative_training_time = (
                total_train_time / self.data_parallel_world_size
            )

        overflow = False
        try:
            with torch.autograd.profiler.record_function("reduce-grads"):
                self.optimizer.all_reduce_grads(self.model)

            with torch.autograd.profiler.record_function("multiply-grads"):
                # grads arrive pre-divided by world size (DDP convention,
                # better low-precision accuracy), so scale by
                # world/sample_size to end at sum_of_gradients/sample_size;
                # for fp16 this also folds in the 1/loss_scale. Some
                # optimizers defer this factor, so raw p.grad may still
                # look unscaled.
                numer = (
                    self.data_parallel_world_size if self._sync_stats() else 1
                )
                self.optimizer.multiply_grads(numer / (sample_size or 1.0))
                # (sample_size or 1.0) guards the all-dummy case

            with torch.autograd.profiler.record_function("clip-grads"):
                grad_norm = self.clip_grad_norm(self.args.clip_norm)

            self._check_grad_norms(grad_norm)
            if not torch.isfinite(grad_norm).all():
                # single-GPU non-finite grads go through the same
                # NanDetector path as the cross-rank check
                raise FloatingPointError("gradients are Nan/Inf")

            with torch.autograd.profiler.record_function("optimizer"):
                # rank-IDENTICAL seed: stochastic rounding must agree
                # across replicas or the weights drift apart
                with utils.torch_seed(self.args.seed, self.get_num_updates()):
                    self.task.optimizer_step(
                        self.optimizer, model=self.model,
                        update_num=self.get_num_updates(),
                    )
            if self.ema is not None:
                with torch.autograd.profiler.record_function("ema"):
                    source = (
                        self.optimizer.fp32_params
                        if self.args.fp16 or self.args.bf16
                        else self._model.named_parameters()
                    )
                    self.ema.update(source)

        except FloatingPointError:
            # forensic re-run with hooks to name the first bad module
            self.zero_grad()
            with NanDetector(self.get_model()):
                for _, sample in enumerate(samples):
                    sample, _ = self._prepare_sample(sample)
                    self.task.train_step(
                        sample, self.model, self.loss, self.optimizer,
                        self.get_num_updates(), ignore_grad=False,
                    )
            raise
        except OverflowError as exc:
            overflow = True
            logger.info(f"NOTE: gradient overflow detected, ignoring"
                        f" gradient, {exc}")
            grad_norm = torch.tensor(0.0)
            if self.cuda:
                grad_norm = grad_norm.cuda()
            self.zero_grad()
        except RuntimeError as exc:
            if _looks_like_oom(exc):
                self._log_oom(exc)
                logger.error("OOM during optimization, irrecoverable")
            raise

        logging_output = None
        if not overflow:
            self.set_num_updates(self.get_num_updates() + 1)
            self._log_memory_headroom()
            logging_output = self._reduce_and_log_stats(
                logging_outputs, sample_size, grad_norm
            )
            self._maybe_empty_cache()

        if self.args.fp16:
            metrics.log_scalar(
                "loss_scale", self.optimizer.scaler.loss_scale,
                priority=700, round=4, weight=0,
            )

        metrics.log_stop_time("train_wall")
        return logging_output

    def _log_memory_headroom(self):
        if not (self.cuda and self.cuda_env is not None):
            return
        gb_used = torch.cuda.max_memory_allocated() / 1024**3
        torch.cuda.reset_peak_memory_stats()
        gb_free = self.cuda_env.total_memory_in_GB - gb_used
        metrics.log_scalar("gb_free", gb_free, priority=1500, round=1,
                           weight=0)

    def _maybe_empty_cache(self):
        """Periodic cache flush against fragmentation (--empty-cache-freq)."""
        freq = self.args.empty_cache_freq
        if not (self.cuda and freq > 0):
            return
        if (self.get_num_updates() + freq - 1) % freq == 0:
            torch.cuda.empty_cache()

    @metrics.aggregate("valid")
    def valid_step(self, sample, raise_oom=False):
        """Forward-only evaluation step (with one OOM retry)."""
        with torch.no_grad():
            self.model.eval()
            self.loss.eval()

            sample, is_filler = self._prepare_sample(sample)
            try:
                _loss, sample_size, logging_output = self.task.valid_step(
                    sample, self.model, self.loss
                )
            except RuntimeError as exc:
                if _looks_like_oom(exc) and not raise_oom:
                    self._log_oom(exc)
                    logger.warning("ran out of memory in validation step,"
                                   " retrying batch")
                    for p in self.model.parameters():
                        p.grad = None  # release anything backward left over
                    if self.cuda:
                        torch.cuda.empty_cache()
                    return self.valid_step(sample, raise_oom=True)
                raise

            logging_outputs = [logging_output]
            if is_filler:
                sample_size = _zeroed_stat(sample_size)

        if self.data_parallel_world_size > 1:
            logging_outputs, (sample_size,) = self._aggregate_logging_outputs(
                logging_outputs, sample_size, ignore=is_filler
            )
        return self._reduce_and_log_stats(logging_outputs, sample_size)

    def zero_grad(self):
        self.optimizer.zero_grad()
        # lazy-DDP hygiene: an aborted backward (OOM retry) can leave
        # autograd-assigned grads on the params; the next backward would
        # accumulate into them. The flat-view engines re-pin at forward
        # time; the lazy engine drops the assigned tensors here instead.
        engine = self._wrapped_model
        if engine is not None and getattr(engine, "lazy", False):
            engine.zero_grad_buffers()

    # -- LR plumbing ---------------------------------------------------------

    def lr_step_begin_epoch(self, epoch):
        self.lr_scheduler.step_begin_epoch(epoch)
        return self.lr_step_update()  # update-count-driven value wins

    def lr_step(self, epoch, val_loss=None):
        self.lr_scheduler.step(epoch, val_loss)
        return self.lr_step_update()  # update-count-driven value wins

    def lr_step_update(self):
        new_lr = self.lr_scheduler.step_update(self.get_num_updates())
        if not isinstance(new_lr, dict):
            metrics.log_scalar("lr", new_lr, weight=0, priority=300)
            return new_lr
        for k, v in new_lr.items():
            metrics.log_scalar(f"lr_{k}", v, weight=0, priority=300)
        return new_lr.get("default", next(iter(new_lr.values())))

    def get_lr(self):
        return self.optimizer.get_lr()

    def get_model(self):
        """The bare (never DDP-wrapped) model."""
        return self._model

    def get_loss(self):
        """The bare (never DDP-wrapped) loss."""
        return self._loss

    def get_num_updates(self):
        return self._num_updates

    def set_num_updates(self, num_updates):
        self._num_updates = num_updates
        self.lr_step_update()
        metrics.log_scalar("num_updates", self._num_updates, weight=0,
                           priority=200)

    def clip_grad_norm(self, clip_norm):
        return self.optimizer.clip_grad_norm(clip_norm, aggregate_norm_fn=None)

    def cumulative_training_time(self):
        if self._cumulative_training_time is not None:
            return self._cumulative_training_time  # cross-rank average
        return self._local_cumulative_training_time()

    def _local_cumulative_training_time(self):
        return time.time() - self._start_time + self._previous_training_time

    # -- sample prep / seeding ----------------------------------------------

    def _prepare_sample(self, sample, is_dummy=False):
        if sample == "DUMMY":
            raise Exception(
                "Trying to use an uninitialized 'dummy' batch. This usually "
                "indicates that the total number of batches is smaller than "
                "the number of participating GPUs. Try reducing the batch "
                "size or using fewer GPUs."
            )

        if sample is None or len(sample) == 0:
            assert (self._dummy_batch is not None
                    and len(self._dummy_batch) > 0), \
                f"Invalid dummy batch: {self._dummy_batch}"
            filled, _ = self._prepare_sample(self._dummy_batch, is_dummy=True)
            return filled, True

        if self.cuda:
            sample = utils.move_to_cuda(sample)

        if self.args.fp16 or self.args.bf16:
            target = torch.half if self.args.fp16 else torch.bfloat16
            sample = utils.apply_to_sample(
                lambda t: t.to(target) if t.dtype is torch.float32 else t,
                sample,
            )

        if self._dummy_batch == "DUMMY":
            self._dummy_batch = sample
        return sample, False

    def _set_seed(self):
        # keyed by update count so resumed runs replay the same stream
        seed = self.args.seed + self.get_num_updates()
        torch.manual_seed(seed)
        if self.cuda:
            torch.cuda.manual_seed(seed)

    def _sync_stats(self):
        return self.data_parallel_world_size > 1

    def _log_oom(self, exc):
        logger.warning(f"OOM: Ran out of memory with exception: {exc}")
        if self.cuda and hasattr(torch.cuda, "memory_summary"):
            for device_idx in range(torch.cuda.device_count()):
                logger.warning(torch.cuda.memory_summary(device=device_idx))
        sys.stderr.flush()

    # -- cross-rank stat sync -------------------------------------------------

    def _aggregate_logging_outputs(self, logging_outputs, *extra_stats_to_sum,
                                   ignore=False):
        summable = self.task.__class__.logging_outputs_can_be_summed(
            self.get_loss(), is_train=self.model.training
        )
        sync = self._fast_stat_sync_sum if summable else self._all_gather_list_sync
        return sync(logging_outputs, *extra_stats_to_sum, ignore=ignore)

    def _all_gather_list_sync(self, logging_outputs, *extra_stats_to_sum,
                              ignore=False):
        """Pickle-everything sync: handles arbitrarily structured logging
        outputs at the cost of serialization."""
        if ignore:
            logging_outputs = list()
        gathered = list(zip(*distributed_utils.all_gather_list(
            [logging_outputs] + list(extra_stats_to_sum),
            max_size=getattr(self.args, "all_gather_list_size", 16384),
            group=self.data_parallel_process_group,
        )))
        per_rank_logs, per_rank_extras = gathered[0], gathered[1:]
        return (
            list(chain.from_iterable(per_rank_logs)),
            [sum(stat) for stat in per_rank_extras],
        )

    def _fast_stat_sync_sum(self, logging_outputs, *extra_stats_to_sum,
                            ignore=False):
        """One concatenated all-reduce of plain scalars — requires flat,
        summable logging outputs (no nesting)."""
        bundle = {
            f"extra_stats_{i}": stat
            for i, stat in enumerate(extra_stats_to_sum)
        }
        log_keys = None
        if logging_outputs:
            log_keys = list(logging_outputs[0].keys())
            for k in log_keys:
                if ignore:
                    probe = logging_outputs[0][k]
                    value = (
                        torch.zeros_like(probe) if torch.is_tensor(probe)
                        else 0
                    )
                if not ignore:
                    value = sum(log[k] for log in logging_outputs if k in log)
                bundle[f"logging_outputs_{k}"] = value

        bundle = distributed_utils.all_reduce_dict(
            bundle, device=self.device, group=self.data_parallel_process_group
        )

        extras = [
            bundle[f"extra_stats_{i}"]
            for i in range(len(extra_stats_to_sum))
        ]
        logs = (
            [{k: bundle[f"logging_outputs_{k}"] for k in log_keys}]
            if log_keys is not None
            else []
        )
        return logs, extras

    def _check_grad_norms(self, grad_norm):
        """Cross-rank agreement check: replicas must see (nearly) identical
        global grad norms, or the graphs have diverged."""
        if self._grad_norm_buf is None:
            return
        buf = self._grad_norm_buf
        buf.zero_()
        buf[self.data_parallel_rank] = grad_norm
        distributed_utils.all_reduce(
            buf, group=self.data_parallel_process_group
        )

        def agrees(t):
            spread = torch.max(torch.abs(t - t[0]))
            all_close = (
                torch.isfinite(t).all()
                and (spread / (t[0] + 1e-6) < 1e-6).all()
            )
            all_bad = (torch.isnan(t) | torch.isinf(t)).all()
            return all_close or all_bad

        if not agrees(buf):
            table = "\n".join(
                f"rank {r:3d} = {n:.8f}" for r, n in enumerate(buf.tolist())
            )
            # FloatingPointError routes into the NanDetector re-run
            raise FloatingPointError(
                "Fatal error: gradients are inconsistent between workers. "
                "Try --ddp-backend=legacy_ddp. "
                "Or are you mixing up different generation of GPUs in "
                "training?\n"
                + "-" * 80
                + f"\ngrad_norm across the workers:\n{table}\n\n"
                + "-" * 80
            )

    def _reduce_and_log_stats(self, logging_outputs, sample_size,
                              grad_norm=None):
        if grad_norm is not None and (
            not torch.is_tensor(grad_norm) or torch.isfinite(grad_norm)
        ):
            metrics.log_speed("ups", 1.0, priority=100, round=2)
            metrics.log_scalar("gnorm", grad_norm, priority=400, round=3)
            if self.args.clip_norm > 0:
                clipped = torch.where(
                    grad_norm > self.args.clip_norm,
                    grad_norm.new_tensor(100),
                    grad_norm.new_tensor(0),
                )
                metrics.log_scalar("clip", clipped, priority=500, round=1)

        with metrics.aggregate() as agg:
            if logging_outputs is not None:
                self.task.reduce_metrics(logging_outputs, self.get_loss())
                del logging_outputs

            if "loss" not in agg:
                # a loss that never logs "loss" breaks best-checkpoint logic
                if "loss" not in self._warn_once:
                    self._warn_once.add("loss")
                    logger.warning(
                        "Loss.reduce_metrics did not log a 'loss' value, "
                        "which may break some functionality"
                    )
                metrics.log_scalar("loss", -1)

            logging_output = agg.get_smoothed_values()
            logging_output["sample_size"] = sample_size
            for legacy_key in ("ppl", "wps", "wpb", "bsz"):
                logging_output.pop(legacy_key, None)
            return logging_output
