"""Inverse-sqrt LR schedule (parity: reference
unicore/optim/lr_scheduler/inverse_square_root_schedule.py:14-77).

Linear warmup from ``--warmup-init-lr`` to ``--lr`` over
``--warmup-updates`` steps, then ``lr = lr_peak * sqrt(warmup) / sqrt(n)``
(the two phases join continuously at the warmup boundary).
"""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("inverse_sqrt")
class InverseSquareRootSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        if isinstance(args.lr, (list, tuple)) and len(args.lr) > 1:
            raise ValueError(
                "Cannot use a fixed learning rate schedule with inverse_sqrt."
                " Consider --lr-scheduler=fixed instead."
            )
        peak = args.lr[0] if isinstance(args.lr, (list, tuple)) else args.lr
        if args.warmup_init_lr < 0:
            args.warmup_init_lr = 0 if args.warmup_updates > 0 else peak

        # warmup slope, then the constant that makes decay continuous
        self.lr_step = (peak - args.warmup_init_lr) / args.warmup_updates
        self.decay_factor = peak * args.warmup_updates**0.5

        self.lr = args.warmup_init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--warmup-updates", default=4000, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")
        parser.add_argument("--warmup-init-lr", default=-1, type=float,
                            metavar="LR",
                            help="starting warmup LR (defaults from --lr)")

    def step(self, epoch, val_loss=None):
        super().step(epoch, val_loss)
        # purely update-driven: no epoch-boundary change
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        if num_updates < self.args.warmup_updates:
            self.lr = self.args.warmup_init_lr + num_updates * self.lr_step
        else:
            self.lr = self.decay_factor / num_updates**0.5
        self.optimizer.set_lr(self.lr)
        return self.lr
