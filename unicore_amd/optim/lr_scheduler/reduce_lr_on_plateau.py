"""Plateau-driven LR decay (parity: reference
unicore/optim/lr_scheduler/reduce_lr_on_plateau.py:16-116).

Wraps torch's ReduceLROnPlateau on the validation loss at epoch ends, with
an optional linear warmup during which ``step_update`` owns the LR; the
plateau scheduler only engages once warmup completes.
"""

import torch.optim.lr_scheduler

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("reduce_lr_on_plateau")
class ReduceLROnPlateauLRSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        if isinstance(args.lr, (list, tuple)) and len(args.lr) > 1:
            raise ValueError(
                "Cannot use a fixed learning rate schedule with "
                "reduce_lr_on_plateau. Consider --lr-scheduler=fixed instead."
            )
        self.lr_scheduler = torch.optim.lr_scheduler.ReduceLROnPlateau(
            self.optimizer.optimizer,
            patience=args.lr_patience,
            factor=args.lr_shrink,
            mode="max" if args.maximize_best_checkpoint_metric else "min",
            threshold=args.lr_threshold,
        )
        peak = args.lr[0]
        if args.warmup_init_lr < 0:
            args.warmup_init_lr = 0 if args.warmup_updates > 0 else peak
        if args.warmup_updates > 0:
            self.warmup_factor = 1.0 / args.warmup_updates

        # toggled by step_update() once the ramp completes (or immediately
        # when there is no warmup)
        self.warmup_end = args.warmup_updates <= 0

        # self.lr only matters at init / during warmup
        self.lr = peak if self.warmup_end else args.warmup_init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--lr-shrink", default=0.1, type=float,
                            metavar="LS",
                            help="plateau shrink factor, lr_new = lr * lr_shrink")
        parser.add_argument("--lr-threshold", default=1e-4, type=float,
                            metavar="LT",
                            help="minimum improvement that resets the plateau")
        parser.add_argument("--lr-patience", default=0, type=int,
                            help="epochs without improvement before decaying")
        parser.add_argument("--warmup-updates", default=0, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")
        parser.add_argument("--warmup-init-lr", default=-1, type=float,
                            metavar="LR",
                            help="starting warmup LR (defaults from --lr)")

    def state_dict(self) -> dict:
        return {
            "best": self.lr_scheduler.best,
            "last_epoch": self.lr_scheduler.last_epoch,
        }

    def load_state_dict(self, state: dict) -> None:
        self.lr_scheduler.best = state["best"]
        if "last_epoch" in state:
            self.lr_scheduler.last_epoch = state["last_epoch"]

    def step(self, epoch, val_loss=None):
        """Feed the epoch's validation loss to the plateau logic (only after
        warmup has finished)."""
        if val_loss is not None and self.warmup_end:
            self.lr_scheduler.step(val_loss)
        else:
            self.lr_scheduler.last_epoch = epoch
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        """Drive the LR through the warmup ramp; hand off afterwards."""
        warmup = self.args.warmup_updates
        if warmup > 0:
            if num_updates <= warmup:
                self.warmup_factor = num_updates / float(warmup)
                start = self.args.warmup_init_lr
                self.optimizer.set_lr(
                    start + self.warmup_factor * (self.args.lr[0] - start)
                )
            elif not self.warmup_end:
                self.warmup_end = True
        return self.optimizer.get_lr()
