"""Exponential decay LR schedule (parity: reference
unicore/optim/lr_scheduler/exponential_decay_schedule.py:12-50).

After a linear warmup, ``lr = lr0 * decay_ratio^(n / decay_steps)`` —
continuous by default, staircase (integer exponent) with ``--stair-decay``.
"""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("exponential_decay")
class ExponentialDecaySchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        self.lr = args.lr[0]
        self.warmup_factor = (
            1.0 / args.warmup_updates if args.warmup_updates > 0 else 1
        )
        self.decay_ratio = args.decay_ratio
        self.decay_steps = args.decay_steps
        self.stair_decay = args.stair_decay
        self.optimizer.set_lr(self.warmup_factor * self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--warmup-updates", default=1000, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")
        parser.add_argument("--decay-ratio", default=0.95, type=float,
                            metavar="R",
                            help="multiplier applied per decay-steps updates")
        parser.add_argument("--decay-steps", default=500, type=int,
                            metavar="N", help="updates per decay application")
        parser.add_argument("--stair-decay", action="store_true",
                            help="integer (staircase) decay exponents")

    def step_update(self, num_updates):
        warmup = self.args.warmup_updates
        if warmup > 0 and num_updates < warmup:
            self.warmup_factor = (num_updates + 1) / float(warmup)
            lr = self.warmup_factor * self.lr
        else:
            exponent = num_updates / self.decay_steps
            if self.stair_decay:
                exponent = int(exponent)
            lr = self.lr * self.decay_ratio**exponent
        self.optimizer.set_lr(lr)
        return self.optimizer.get_lr()
