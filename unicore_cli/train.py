#!/usr/bin/env python3 -u
"""unicore-train entrypoint: epoch loop, validate-and-save, early stopping.

Parity: reference unicore_cli/train.py — main:43, train:178,
validate_and_save:251, validate:337, should_stop_early:149, cli_main:409.

Control flow: ``cli_main`` parses and dispatches through the distributed
launcher into ``main``, which builds task/model/loss/Trainer, restores the
latest checkpoint, and runs ``train`` once per epoch; ``validate_and_save``
evaluates the stop conditions (max_update, wall-clock budget, min LR,
patience) after every update group.
"""

import logging
import math
import os
import random
import sys
from multiprocessing.pool import ThreadPool

import numpy as np
import torch

from unicore_amd import checkpoint_utils, options, tasks, utils
from unicore_amd.data import iterators
from unicore_amd.distributed import utils as distributed_utils
from unicore_amd.logging import meters, metrics, progress_bar
from unicore_amd.trainer import Trainer

logging.basicConfig(
    format="%(asctime)s | %(levelname)s | %(name)s | %(message)s",
    datefmt="%Y-%m-%d %H:%M:%S",
    level=os.environ.get("LOGLEVEL", "INFO").upper(),
    stream=sys.stdout,
)
logger = logging.getLogger("unicore_cli.train")


def _seed_everything(seed):
    torch.manual_seed(seed)
    np.random.seed(seed)
    random.seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)


def _count_params(model, trainable_only=False):
    return sum(
        getattr(p, "_orig_size", p).numel()
        for p in model.parameters()
        if p.requires_grad or not trainable_only
    )


def main(args) -> None:
    utils.import_user_module(args)
    utils.set_jit_fusion_options()
    if getattr(args, "gemm_tuning_file", None):
        if utils.load_gemm_tunings(args.gemm_tuning_file):
            logger.info(f"loaded GEMM tunings from {args.gemm_tuning_file}")

    assert args.batch_size is not None, \
        "Must specify batch size either with --batch-size"

    metrics.reset()
    # clear any best-metric state from a previous in-process run; a resumed
    # run restores it from the checkpoint's extra_state instead
    for holder, attr in ((checkpoint_utils.save_checkpoint, "best"),
                         (should_stop_early, "best")):
        if hasattr(holder, attr):
            delattr(holder, attr)
    should_stop_early.num_runs = 0

    _seed_everything(args.seed)

    if distributed_utils.is_master(args):
        checkpoint_utils.verify_checkpoint_directory(args.save_dir)

    logger.info(args)

    task = tasks.setup_task(args)
    assert args.loss, "Please specify loss to train a model"

    model = task.build_model(args)
    loss = task.build_loss(args)
    logger.info(model)
    logger.info(f"task: {type(task).__name__}")
    logger.info(f"model: {type(model).__name__}")
    logger.info(f"loss: {type(loss).__name__}")
    logger.info(
        f"num. model params: {_count_params(model):,} "
        f"(num. trained: {_count_params(model, trainable_only=True):,})"
    )

    # validation data loads now; training data is tied to the checkpoint's
    # iterator position and loads inside load_checkpoint
    for valid_sub_split in args.valid_subset.split(","):
        task.load_dataset(valid_sub_split, combine=False, epoch=1)

    trainer = Trainer(args, task, model, loss)
    logger.info(f"training on {args.distributed_world_size} devices (GPUs)")
    logger.info(f"batch size per device = {args.batch_size}")

    extra_state, epoch_itr = checkpoint_utils.load_checkpoint(
        args, trainer, disable_iterator_cache=True
    )

    max_epoch = args.max_epoch if args.max_epoch else math.inf
    learn_rate = trainer.get_lr()
    wall_timer = meters.StopwatchMeter()
    wall_timer.start()
    ckp_copy_thread = ThreadPool(1)

    while epoch_itr.next_epoch_idx <= max_epoch:
        if learn_rate <= args.stop_min_lr:
            logger.info(
                f"stopping training because current learning rate "
                f"({learn_rate}) is smaller than or equal to minimum "
                f"learning rate (--stop-min-lr={args.stop_min_lr})"
            )
            break

        valid_losses, should_stop = train(args, trainer, task, epoch_itr,
                                          ckp_copy_thread)
        if should_stop:
            break

        # the first validation subset drives the LR schedule
        learn_rate = trainer.lr_step(epoch_itr.epoch, valid_losses[0])

        epoch_itr = trainer.get_train_iterator(
            epoch_itr.next_epoch_idx,
            # sharded data rotates files between epochs
            load_dataset=task.has_sharded_data("train"),
            disable_iterator_cache=True,
        )

    ckp_copy_thread.close()
    ckp_copy_thread.join()
    wall_timer.stop()
    logger.info(f"done training in {wall_timer.sum:.1f} seconds")


def should_stop_early(args, valid_loss: float) -> bool:
    """Patience-based early stop, tracking the best loss on the function
    object (cleared per in-process run)."""
    if valid_loss is None or args.patience <= 0:
        return False

    def improved(a, b):
        if args.maximize_best_checkpoint_metric:
            return a > b
        return a < b

    best_so_far = getattr(should_stop_early, "best", None)
    if best_so_far is None or improved(valid_loss, best_so_far):
        should_stop_early.best, should_stop_early.num_runs = valid_loss, 0
        return False
    should_stop_early.num_runs += 1
    out_of_patience = should_stop_early.num_runs >= args.patience
    if out_of_patience:
        logger.info(
            "early stop since valid performance hasn't improved for last "
            f"{args.patience} runs"
        )
    return out_of_patience


def _epoch_update_freq(args, epoch):
    schedule = args.update_freq
    return schedule[epoch - 1] if epoch <= len(schedule) else schedule[-1]


def _build_progress(args, itr, epoch, prefix=None):
    on_master = distributed_utils.is_master(args)
    return progress_bar.progress_bar(
        itr,
        log_format=args.log_format, log_interval=args.log_interval,
        epoch=epoch,
        prefix=prefix,
        tensorboard_logdir=args.tensorboard_logdir if on_master else None,
        default_log_format="simple" if args.no_progress_bar else "tqdm",
        wandb_project=(
            args.wandb_project if on_master and args.wandb_project else None
        ),
        args=args,
    )


@metrics.aggregate("train")
def train(args, trainer, task, epoch_itr, ckp_copy_thread):
    """One epoch of updates; returns (valid_losses, should_stop)."""
    itr = epoch_itr.next_epoch_itr(
        fix_batches_to_gpus=args.fix_batches_to_gpus,
        shuffle=(epoch_itr.next_epoch_idx > args.curriculum),
    )
    itr = iterators.GroupedIterator(
        itr, _epoch_update_freq(args, epoch_itr.epoch)
    )
    progress = _build_progress(args, itr, epoch_itr.epoch)

    trainer.begin_epoch(epoch_itr.epoch)

    valid_subsets = args.valid_subset.split(",")
    valid_losses, should_stop = [None], False
    num_updates = trainer.get_num_updates()
    cap = args.max_update if args.max_update else math.inf
    logger.info("Start iterating over samples")
    for i, samples in enumerate(progress):
        with metrics.aggregate("train_inner"), \
                torch.autograd.profiler.record_function("train_step-%d" % i):
            log_output = trainer.train_step(samples)

        if log_output is not None:  # skipped steps (OOM/overflow) log nothing
            num_updates = trainer.get_num_updates()
            if num_updates % args.log_interval == 0:
                stats = get_training_stats(
                    metrics.get_smoothed_values("train_inner")
                )
                progress.log(stats, tag="train_inner", step=num_updates)
                # restart the mid-epoch window (epoch-level meters persist)
                metrics.reset_meters("train_inner")

        end_of_epoch = not itr.has_next()
        valid_losses, should_stop = validate_and_save(
            args, trainer, task, epoch_itr, valid_subsets, end_of_epoch,
            ckp_copy_thread,
        )
        if should_stop or num_updates >= cap:
            should_stop = True
            break

    logger.info(
        f"end of epoch {epoch_itr.epoch} (average epoch stats below)"
    )
    stats = get_training_stats(metrics.get_smoothed_values("train"))
    progress.print(stats, tag="train", step=num_updates)

    metrics.reset_meters("train")
    return valid_losses, should_stop


def validate_and_save(args, trainer, task, epoch_itr, valid_subsets,
                      end_of_epoch, ckp_copy_thread):
    """Evaluate stop conditions, then validate/checkpoint as scheduled."""
    num_updates, cap = trainer.get_num_updates(), (
        args.max_update if args.max_update else math.inf
    )

    should_stop = num_updates >= cap
    if should_stop:
        logger.info(f"Stopping training due to num_updates: {num_updates}"
                    f" >= max_update: {cap}")

    trained_hours = trainer.cumulative_training_time() / 3600.0
    if 0 < args.stop_time_hours < trained_hours:
        should_stop = True
        logger.info(f"Stopping training due to cumulative_training_time:"
                    f" {trained_hours} > stop_time_hours:"
                    f" {args.stop_time_hours} hour(s)")

    hit_save_interval = (
        args.save_interval_updates > 0 < num_updates
        and num_updates % args.save_interval_updates == 0
        and num_updates >= args.validate_after_updates
    )
    do_save = (
        should_stop
        or (end_of_epoch and epoch_itr.epoch % args.save_interval == 0)
        or hit_save_interval
    )
    hit_valid_interval = (
        args.validate_interval_updates > 0 < num_updates
        and num_updates % args.validate_interval_updates == 0
    )
    do_validate = not args.disable_validation and (
        should_stop
        or (not end_of_epoch and do_save)  # mid-epoch saves validate too
        or (end_of_epoch and epoch_itr.epoch % args.validate_interval == 0)
        or hit_valid_interval
    )

    valid_losses = [None]
    if do_validate:
        with utils.validate_with_ema(trainer, ema=args.validate_with_ema):
            valid_losses = validate(
                args, trainer, task, epoch_itr, valid_subsets
            )

    should_stop |= should_stop_early(args, valid_losses[0])

    if do_save or should_stop:
        checkpoint_utils.save_checkpoint(
            args, trainer, epoch_itr, valid_losses[0], ckp_copy_thread
        )
    return (valid_losses, should_stop)


def get_training_stats(stats):
    stats["wall"] = round(metrics.get_meter("default", "wall").elapsed_time, 0)
    return stats


def validate(args, trainer, task, epoch_itr, subsets):
    """Run every validation subset; returns the best-metric value per
    subset."""
    seed = args.fixed_validation_seed  # None leaves the RNG stream alone

    with utils.torch_seed(seed):
        trainer.begin_valid_epoch(epoch_itr.epoch)
        valid_losses = []
        for subset in subsets:
            logger.info(f'begin validation on "{subset}" subset')

            itr = trainer.get_valid_iterator(subset).next_epoch_itr(
                shuffle=False
            )
            progress = _build_progress(
                args, itr, epoch_itr.epoch,
                prefix=f"valid on '{subset}' subset",
            )

            # fresh root aggregator: validation numbers never leak into the
            # train meters
            with metrics.aggregate(new_root=True) as agg:
                for i, sample in enumerate(progress):
                    if (args.max_valid_steps is not None
                            and i > args.max_valid_steps):
                        break
                    trainer.valid_step(sample)

            stats = get_valid_stats(args, trainer, agg.get_smoothed_values())
            progress.print(stats, tag=subset, step=trainer.get_num_updates())
            valid_losses.append(stats[args.best_checkpoint_metric])
    return valid_losses


def get_valid_stats(args, trainer, stats):
    stats["num_updates"] = trainer.get_num_updates()
    if hasattr(checkpoint_utils.save_checkpoint, "best"):
        pick = max if args.maximize_best_checkpoint_metric else min
        stats[f"best_{args.best_checkpoint_metric}"] = pick(
            checkpoint_utils.save_checkpoint.best,
            stats[args.best_checkpoint_metric],
        )
    return stats


def cli_main(modify_parser=None) -> None:
    parser = options.get_training_parser()
    args = options.parse_args_and_arch(parser, modify_parser=modify_parser)

    launch = distributed_utils.call_main
    if args.profile:
        # emit rocTX ranges for every autograd op under rocprof
        with torch.cuda.profiler.profile(), \
                torch.autograd.profiler.emit_nvtx():
            launch(args, main)
    else:
        launch(args, main)


if __name__ == "__main__":
    cli_main()
