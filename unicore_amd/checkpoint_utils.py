"""Checkpoint save/load in the Uni-Core checkpoint_utils format.

Parity: reference unicore/checkpoint_utils.py — save_checkpoint:83,
load_checkpoint:165, load_checkpoint_to_cpu:244, checkpoint_paths:261,
torch_persistent_save:280, async copy/prune thread ckp_copy_fun:23-80.
File schema per SURVEY.md Appendix B (single torch.save dict with args,
model, loss, optimizer_history, task_state, extra_state,
last_optimizer_state, ema).

Write path: rank 0 serializes ONCE into tmp_save_dir (atomic tmp+rename),
then the async copy thread fans the file out to every active checkpoint
name and applies the keep policies — the training loop never blocks on
filesystem copies.
"""

import collections
import logging
import os
import re
import shutil
import traceback

import torch

logger = logging.getLogger(__name__)

_SAVE_RETRIES = 3


def _prune(paths):
    for stale in paths:
        if os.path.lexists(stale):
            os.remove(stale)
            logger.info(f"removed {stale}")


def ckp_copy_fun(src, checkpoints, end_of_epoch, args):
    """Async-thread body: fan the staged checkpoint out to its final names,
    then enforce the retention policies."""
    copied_any = False
    src_is_staging = src != checkpoints[0]
    try:
        for destination in checkpoints:
            if destination != src:
                logger.info(f"copy {src} to {destination}")
                shutil.copyfile(src, destination)
                copied_any = True
        if src_is_staging and copied_any and os.path.lexists(src):
            os.remove(src)

        sfx = args.checkpoint_suffix
        if not end_of_epoch and args.keep_interval_updates > 0:
            # update checkpoints beyond the newest N (list sorts descending)
            stale = checkpoint_paths(
                args.save_dir,
                pattern=rf"checkpoint_\d+_(\d+){sfx}\.pt",
            )[args.keep_interval_updates:]
            _prune(stale)

        if args.keep_last_epochs > 0:
            stale = checkpoint_paths(
                args.save_dir,
                pattern=rf"checkpoint(\d+){sfx}\.pt",
            )[args.keep_last_epochs:]
            _prune(stale)

        if args.keep_best_checkpoints > 0:
            # by-metric retention: order so the best sort first, drop the rest
            ranked = checkpoint_paths(
                args.save_dir,
                pattern=rf"checkpoint\.best_{args.best_checkpoint_metric}"
                        rf"_(\d+\.?\d*){sfx}\.pt",
            )
            if not args.maximize_best_checkpoint_metric:
                ranked = ranked[::-1]
            _prune(ranked[args.keep_best_checkpoints:])
    except Exception:
        logger.error(
            f"Error in checkpoint copy/prune thread: {traceback.format_exc()}"
        )


def is_better(args, a, b):
    return a >= b if args.maximize_best_checkpoint_metric else a <= b


def _active_checkpoint_names(args, epoch, end_of_epoch, updates, val_loss,
                             suffix):
    """Which checkpoint filenames this save should produce."""
    conds = collections.OrderedDict()
    conds[f"checkpoint{epoch}{suffix}.pt"] = (
        end_of_epoch
        and not args.no_epoch_checkpoints
        and epoch % args.save_interval == 0
    )
    conds[f"checkpoint_{epoch}_{updates}{suffix}.pt"] = (
        not end_of_epoch
        and args.save_interval_updates > 0
        and updates % args.save_interval_updates == 0
    )
    improved = val_loss is not None and (
        not hasattr(save_checkpoint, "best")
        or is_better(args, val_loss, save_checkpoint.best)
    )
    conds[f"checkpoint_best{suffix}.pt"] = improved
    if val_loss is not None and args.keep_best_checkpoints > 0:
        name = (
            f"checkpoint.best_{args.best_checkpoint_metric}_{val_loss:.2f}.pt"
        )
        conds[name] = improved
    conds[f"checkpoint_last{suffix}.pt"] = not args.no_last_checkpoints
    return [fn for fn, keep in conds.items() if keep]


def save_checkpoint(args, trainer, epoch_itr, val_loss, ckp_copy_thread,
                    do_save=True):
    from unicore_amd.logging import meters

    # a single rank creates the directory
    if trainer.data_parallel_rank == 0:
        os.makedirs(args.save_dir, exist_ok=True)

    prev_best = getattr(save_checkpoint, "best", val_loss)
    if val_loss is not None:
        pick = max if args.maximize_best_checkpoint_metric else min
        save_checkpoint.best = pick(val_loss, prev_best)

    if args.no_save or not do_save:
        return
    if not trainer.should_save_checkpoint_on_current_rank:
        return

    write_timer = meters.StopwatchMeter()
    write_timer.start()

    epoch = epoch_itr.epoch
    end_of_epoch = epoch_itr.end_of_epoch()
    updates = trainer.get_num_updates()
    suffix = args.checkpoint_suffix or ""

    names = _active_checkpoint_names(
        args, epoch, end_of_epoch, updates, val_loss, suffix
    )
    if not names:
        return

    extra_state = {
        "train_iterator": epoch_itr.state_dict(),
        "val_loss": val_loss,
    }
    if hasattr(save_checkpoint, "best"):
        extra_state["best"] = save_checkpoint.best

    targets = [os.path.join(args.save_dir, fn) for fn in names]
    staging_dir = args.tmp_save_dir or "./"
    os.makedirs(staging_dir, exist_ok=True)
    staged = os.path.join(staging_dir, f"checkpoint_last_tmp{suffix}.pt")
    trainer.save_checkpoint(staged, extra_state)

    if ckp_copy_thread is not None:
        ckp_copy_thread.apply_async(
            ckp_copy_fun, (staged, targets, end_of_epoch, args)
        )
    else:
        ckp_copy_fun(staged, targets, end_of_epoch, args)

    write_timer.stop()
    logger.info(
        f"Saved checkpoint {targets[0]} (epoch {epoch} @ {updates} updates, "
        f"score {val_loss}) (writing took {write_timer.sum} seconds)"
    )


def load_checkpoint(args, trainer, **passthrough_args):
    """Resolve the restore path (auto-resume / --restore-file /
    --finetune-from-model), load it through the trainer, and hand back
    (extra_state, epoch_itr). *passthrough_args* reach
    trainer.get_train_iterator."""
    reset_optimizer = args.reset_optimizer
    reset_lr_scheduler = args.reset_lr_scheduler
    reset_meters = args.reset_meters
    reset_dataloader = args.reset_dataloader
    optimizer_overrides = eval(args.optimizer_overrides)

    any_reset = (reset_optimizer or reset_lr_scheduler or reset_meters
                 or reset_dataloader)
    if args.finetune_from_model is not None and any_reset:
        raise ValueError(
            "--finetune-from-model can not be set together with either "
            "--reset-optimizer or reset_lr_scheduler or reset_meters or "
            "reset_dataloader"
        )

    suffix = args.checkpoint_suffix
    if args.restore_file == "checkpoint_last.pt":
        # the auto-resume default
        checkpoint_path = os.path.join(
            args.save_dir, f"checkpoint_last{suffix}.pt"
        )
        if args.finetune_from_model is not None \
                and not os.path.exists(checkpoint_path):
            # nothing to resume: start the finetune from the pretrained
            # weights with everything else fresh
            checkpoint_path = args.finetune_from_model
            reset_optimizer = reset_lr_scheduler = True
            reset_meters = reset_dataloader = True
            logger.info(
                f"loading pretrained model from {checkpoint_path}: "
                "optimizer, lr scheduler, meters, dataloader will be reset"
            )
    elif suffix is not None:
        checkpoint_path = args.restore_file.replace(".pt", suffix + ".pt")
    else:
        checkpoint_path = args.restore_file

    if args.restore_file != "checkpoint_last.pt" and args.finetune_from_model:
        raise ValueError(
            "--finetune-from-model and --restore-file (non-default value) "
            "can not be specified together: " + str(args)
        )

    extra_state, epoch_itr = trainer.load_checkpoint(
        checkpoint_path,
        reset_optimizer,
        reset_lr_scheduler,
        optimizer_overrides,
        reset_meters=reset_meters,
        reset_dataloader=reset_dataloader,
        **passthrough_args,
    )

    keep_best = (
        extra_state is not None
        and "best" in extra_state
        and not reset_optimizer
        and not reset_meters
    )
    if keep_best:
        save_checkpoint.best = extra_state["best"]
    return extra_state, epoch_itr


def load_checkpoint_to_cpu(path, arg_overrides=None):
    """Deserialize a checkpoint onto the CPU, applying arg overrides."""
    with open(path, "rb") as f:
        state = torch.load(
            f, map_location=torch.device("cpu"), weights_only=False
        )
    if arg_overrides is not None and state.get("args") is not None:
        for key, value in arg_overrides.items():
            setattr(state["args"], key, value)
    return state


def checkpoint_paths(path, pattern=r"checkpoint(\d+)\.pt"):
    """All checkpoints under *path* whose names fullmatch *pattern*,
    sorted descending by the first captured group (or listing order)."""
    matcher = re.compile(pattern)
    found = []
    for i, name in enumerate(os.listdir(path)):
        m = matcher.fullmatch(name)
        if m is None:
            continue
        rank = float(m.group(1)) if m.groups() else i
        found.append((rank, name))
    return [os.path.join(path, name) for _, name in sorted(found, reverse=True)]


def torch_persistent_save(obj, filename):
    """Atomic persist: write <name>.tmp, rename over <name>; retried."""
    if not isinstance(filename, str):
        return _torch_persistent_save(obj, filename)
    staged = filename + ".tmp"
    for attempt in range(_SAVE_RETRIES):
        try:
            with open(staged, "wb") as f:
                _torch_persistent_save(obj, f)
            os.rename(staged, filename)
            return
        except Exception:
            if attempt == _SAVE_RETRIES - 1:
                logger.error(traceback.format_exc())
                raise


def _torch_persistent_save(obj, f):
    if isinstance(f, str):
        with open(f, "wb") as handle:
            torch_persistent_save(obj, handle)
        return
    for attempt in range(_SAVE_RETRIES):
        try:
            return torch.save(obj, f)
        except Exception:
            if attempt == _SAVE_RETRIES - 1:
                logger.error(traceback.format_exc())
                raise


def verify_checkpoint_directory(save_dir: str) -> None:
    """Fail fast when the checkpoint directory is not writable."""
    if not os.path.exists(save_dir):
        os.makedirs(save_dir, exist_ok=True)
    probe = os.path.join(save_dir, "dummy")
    try:
        with open(probe, "w"):
            pass
    except OSError as e:
        logger.warning(
            f"Unable to access checkpoint save directory: {save_dir}"
        )
        raise e
    else:
        os.remove(probe)
