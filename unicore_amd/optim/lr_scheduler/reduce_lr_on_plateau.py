"""Reduce-LR-on-plateau schedule (parity: reference
unicore/optim/lr_scheduler/reduce_lr_on_plateau.py:16-116)."""

import torch.optim.lr_scheduler

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("reduce_lr_on_plateau")
class ReduceLROnPlateauLRSchedule(UnicoreLRScheduler):
    """
    Decay the LR by a factor every time the validation loss plateaus, with an
    optional linear warmup phase. During warmup, ``step_update`` drives the lr;
    afterwards the wrapped torch ReduceLROnPlateau takes over at epoch ends.
    """

    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        if isinstance(args.lr, (list, tuple)) and len(args.lr) > 1:
            raise ValueError(
                "Cannot use a fixed learning rate schedule with reduce_lr_on_plateau."
                " Consider --lr-scheduler=fixed instead."
            )
        self.lr_scheduler = torch.optim.lr_scheduler.ReduceLROnPlateau(
            self.optimizer.optimizer,
            patience=args.lr_patience,
            factor=args.lr_shrink,
            mode="max" if args.maximize_best_checkpoint_metric else "min",
            threshold=args.lr_threshold,
        )
        warmup_end_lr = args.lr[0]
        # if no warm up, sets initial lr to be args.lr[0]
        if args.warmup_init_lr < 0:
            args.warmup_init_lr = 0 if args.warmup_updates > 0 else warmup_end_lr

        # linearly warmup for the first args.warmup_updates
        if args.warmup_updates > 0:
            self.warmup_factor = 1.0 / args.warmup_updates

        # this flag is either set from arguments in case no warm up is needed,
        # or set by step_update() when the warmup period is over
        self.warmup_end = True if args.warmup_updates <= 0 else False

        # initial learning rate
        # this self.lr is used only during init and/or warm up period
        self.lr = warmup_end_lr if self.warmup_end else args.warmup_init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        """Add arguments to the parser for this LR scheduler."""
        parser.add_argument(
            "--lr-shrink",
            default=0.1,
            type=float,
            metavar="LS",
            help="shrink factor for annealing, lr_new = (lr * lr_shrink)",
        )
        parser.add_argument(
            "--lr-threshold",
            default=1e-4,
            type=float,
            metavar="LT",
            help="threshold for measuring the new optimum, to only focus on "
            "significant changes",
        )
        parser.add_argument(
            "--lr-patience",
            default=0,
            type=int,
            help="number of epochs with no improvement after which learning "
            "rate will be reduced",
        )
        parser.add_argument(
            "--warmup-updates",
            default=0,
            type=int,
            metavar="N",
            help="warmup the learning rate linearly for the first N updates",
        )
        parser.add_argument(
            "--warmup-init-lr",
            default=-1,
            type=float,
            metavar="LR",
            help="initial learning rate during warmup phase; default is args.lr",
        )

    def state_dict(self):
        """Return the LR scheduler state dict."""
        return {
            "best": self.lr_scheduler.best,
            "last_epoch": self.lr_scheduler.last_epoch,
        }

    def load_state_dict(self, state_dict):
        """Load an LR scheduler state dict."""
        self.lr_scheduler.best = state_dict["best"]
        if "last_epoch" in state_dict:
            self.lr_scheduler.last_epoch = state_dict["last_epoch"]

    def step(self, epoch, val_loss=None):
        """
        Update the learning rate at the end of the given epoch if warmup
        finishes; otherwise no update of lr on epoch boundaries
        """
        if val_loss is not None and self.warmup_end is True:
            self.lr_scheduler.step(val_loss)
        else:
            self.lr_scheduler.last_epoch = epoch
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        """
        Update the learning rate after each update."""
        # if there is warmup
        if self.args.warmup_updates > 0:
            if num_updates <= self.args.warmup_updates:
                self.warmup_factor = num_updates / float(self.args.warmup_updates)
                lr = self.args.warmup_init_lr + self.warmup_factor * (
                    self.args.lr[0] - self.args.warmup_init_lr
                )
                self.optimizer.set_lr(lr)
            else:
                if self.warmup_end is False:
                    self.warmup_end = True
        # else do nothing
        return self.optimizer.get_lr()
