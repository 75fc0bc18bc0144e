"""Exponential decay LR schedule (parity: reference
unicore/optim/lr_scheduler/exponential_decay_schedule.py:12-50)."""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("exponential_decay")
class ExponentialDecaySchedule(UnicoreLRScheduler):
    """Decay the LR by a multiplicative ratio every fixed number of steps,
    with linear warmup."""

    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        self.lr = args.lr[0]
        if args.warmup_updates > 0:
            self.warmup_factor = 1.0 / args.warmup_updates
        else:
            self.warmup_factor = 1
        self.decay_ratio = args.decay_ratio
        self.decay_steps = args.decay_steps
        self.stair_decay = args.stair_decay
        self.optimizer.set_lr(self.warmup_factor * self.lr)

    @classmethod
    def add_args(cls, parser):
        """Add arguments to the parser for this LR scheduler."""
        parser.add_argument(
            "--warmup-updates",
            default=1000,
            type=int,
            metavar="N",
            help="warmup the learning rate linearly for the first N updates",
        )
        parser.add_argument(
            "--decay-ratio",
            default=0.95,
            type=float,
            metavar="R",
            help="decay ratio per decay-steps updates",
        )
        parser.add_argument(
            "--decay-steps",
            default=500,
            type=int,
            metavar="N",
            help="number of updates per decay",
        )
        parser.add_argument(
            "--stair-decay",
            action="store_true",
            help="use staircase (integer) decay exponents",
        )

    def step_update(self, num_updates):
        """Update the learning rate after each update."""
        if self.args.warmup_updates > 0 and num_updates < self.args.warmup_updates:
            self.warmup_factor = (num_updates + 1) / float(self.args.warmup_updates)
            lr = self.warmup_factor * self.lr
        else:
            exponent = num_updates / self.decay_steps
            if self.stair_decay:
                exponent = int(exponent)
            lr = self.lr * (self.decay_ratio**exponent)
        self.optimizer.set_lr(lr)
        return self.optimizer.get_lr()
