"""Triangular (cyclic) LR schedule (parity: reference
unicore/optim/lr_scheduler/triangular_lr_scheduler.py:14-76; CLR,
https://arxiv.org/pdf/1506.01186.pdf).

Ramps linearly between ``--lr`` (floor) and ``--max-lr`` (ceiling) over a
cycle of ``--lr-period-updates`` steps; every completed cycle shrinks the
ceiling (and optionally the floor) by ``--lr-shrink``.
"""

import math

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("triangular")
class TriangularLRSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        if isinstance(args.lr, (list, tuple)) and len(args.lr) > 1:
            raise ValueError(
                "Cannot use a fixed learning rate schedule with triangular."
                " Consider --lr-scheduler=fixed instead."
            )

        floor = args.lr[0]
        assert args.max_lr > floor, "max_lr must be more than lr"
        self.min_lr = floor
        self.max_lr = args.max_lr
        self.stepsize = args.lr_period_updates // 2  # half-cycle length
        self.lr_shrink = args.lr_shrink
        self.shrink_min = args.shrink_min

        self.lr = self.min_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--max-lr", required=True, type=float,
                            metavar="LR",
                            help="cycle ceiling (must exceed args.lr)")
        parser.add_argument("--lr-period-updates", default=5000, type=float,
                            metavar="LR", help="updates per full cycle")
        parser.add_argument("--lr-shrink", default=0.1, type=float,
                            metavar="LS", help="per-cycle shrink factor")
        parser.add_argument("--shrink-min", action="store_true",
                            help="also shrink the cycle floor")

    def step(self, epoch, val_loss=None):
        super().step(epoch, val_loss)
        # update-driven schedule: epoch boundaries change nothing
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        cycle = math.floor(num_updates / (2 * self.stepsize))
        shrink = self.lr_shrink**cycle
        hi = self.max_lr * shrink
        lo = self.min_lr * shrink if self.shrink_min else self.min_lr

        # triangle wave in [0, 1]: peak mid-cycle
        x = abs(num_updates / self.stepsize - 2 * (cycle + 1) + 1)
        self.lr = lo + (hi - lo) * max(0, 1 - x)

        self.optimizer.set_lr(self.lr)
        return self.lr
