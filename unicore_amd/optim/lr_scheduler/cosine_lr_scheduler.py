"""Cosine LR schedule with warmup and warm restarts (parity: reference
unicore/optim/lr_scheduler/cosine_lr_scheduler.py:15-140; SGDR,
https://arxiv.org/pdf/1608.03983.pdf).

Linear warmup from ``--warmup-init-lr`` to the peak LR, then repeated
cosine half-waves between max and min. ``--t-mult`` stretches each
successive period; ``--lr-shrink`` scales both endpoints down per restart.
A single-period schedule clamps at its final value instead of restarting.
"""

import math

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("cosine")
class CosineLRSchedule(UnicoreLRScheduler):
    def __init__(self, args, unicore_optimizer, total_train_steps):
        super().__init__(args, unicore_optimizer, total_train_steps)
        if isinstance(args.lr, (list, tuple)) and len(args.lr) > 1:
            raise ValueError(
                "Cannot use a fixed learning rate schedule with cosine."
                " Consider --lr-scheduler=fixed instead."
            )

        self.max_lr = (
            args.lr[0] if isinstance(args.lr, (list, tuple)) else args.lr
        )
        assert self.max_lr > args.min_lr, (
            f"max_lr (={self.max_lr}) must be more than min_lr "
            f"(={args.min_lr})"
        )
        if args.warmup_init_lr < 0:
            args.warmup_init_lr = args.min_lr

        self.warmup_updates = args.warmup_updates
        self.warmup_ratio = args.warmup_ratio
        if self.warmup_ratio > 0:
            assert total_train_steps is not None
            self.warmup_updates = int(self.warmup_ratio * total_train_steps)

        self.t_mult = args.t_mult
        self.period = args.lr_period_updates
        if self.period <= 0:
            assert args.max_update > 0, \
                "Either --max-update or --lr-period-updates must be set"
            self.period = args.max_update - self.warmup_updates

        # warmup slope toward the peak
        self.lr_step = (
            (self.max_lr - args.warmup_init_lr) / self.warmup_updates
            if self.warmup_updates > 0
            else 1
        )
        self.warmup_init_lr = args.warmup_init_lr
        self.lr_shrink = args.lr_shrink

        self.lr = args.warmup_init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--warmup-updates", default=0, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")
        parser.add_argument("--warmup-ratio", default=-1.0, type=float,
                            metavar="N",
                            help="warmup length as a fraction of total steps")
        parser.add_argument("--warmup-init-lr", default=-1, type=float,
                            metavar="LR",
                            help="starting warmup LR (defaults from --lr)")
        parser.add_argument("--min-lr", type=float, default=0.0, metavar="LR",
                            help="cosine floor")
        parser.add_argument("--max-lr", type=float, metavar="LR",
                            help="cosine ceiling (must exceed args.lr)")
        parser.add_argument("--t-mult", default=1, type=float, metavar="LR",
                            help="period growth factor per restart")
        parser.add_argument("--lr-period-updates", default=-1, type=float,
                            metavar="LR", help="length of the first period")
        parser.add_argument("--lr-shrink", default=0.1, type=float,
                            metavar="LS", help="endpoint shrink per restart")

    def step(self, epoch, val_loss=None):
        super().step(epoch, val_loss)
        # update-driven schedule: epoch boundaries change nothing
        return self.optimizer.get_lr()

    def _locate_period(self, steps_past_warmup):
        """Which restart are we in, how long is it, and how far along."""
        if self.t_mult != 1:
            # geometric periods: invert the partial-sum formula
            i = math.floor(math.log(
                1 - steps_past_warmup / self.period * (1 - self.t_mult),
                self.t_mult,
            ))
            length = self.t_mult**i * self.period
            offset = steps_past_warmup - (
                (1 - self.t_mult**i) / (1 - self.t_mult) * self.period
            )
        else:
            i = math.floor(steps_past_warmup / self.period)
            length = self.period
            offset = steps_past_warmup - self.period * i
        return i, length, offset

    def step_update(self, num_updates):
        if num_updates < self.warmup_updates:
            self.lr = self.warmup_init_lr + num_updates * self.lr_step
        else:
            i, length, offset = self._locate_period(
                num_updates - self.warmup_updates
            )
            shrink = self.lr_shrink**i
            lo = self.args.min_lr * shrink
            hi = self.max_lr * shrink
            # single-cycle schedules clamp at the end value
            frac = min(offset / length, 1.0)
            self.lr = lo + 0.5 * (hi - lo) * (1 + math.cos(math.pi * frac))

        self.optimizer.set_lr(self.lr)
        return self.lr
