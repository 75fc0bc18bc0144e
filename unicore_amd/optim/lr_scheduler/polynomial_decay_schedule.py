"""Polynomial decay LR schedule (parity: reference
unicore/optim/lr_scheduler/polynomial_decay_schedule.py:12-79).

Linear warmup, then ``lr = (lr0 - end) * (1 - progress)^power + end`` until
``--total-num-update``, after which the end LR holds. With
``--warmup-ratio > 0`` both the warmup length and the decay horizon are
derived from the run's total step count instead.
"""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("polynomial_decay")
class PolynomialDecaySchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        assert args.total_num_update > 0, \
            "Must set --total-num-update with polynomial_decay scheduler"

        self.lr = args.lr[0]
        self.warmup_updates, self.warmup_ratio = (args.warmup_updates,
                                                  args.warmup_ratio)
        self.total_num_update = args.total_num_update
        if self.warmup_ratio > 0:
            # derive both horizons from the actual run length
            assert total_train_steps is not None, "need the run's step count"
            self.total_num_update = int(total_train_steps)
            self.warmup_updates = int(self.warmup_ratio * total_train_steps)

        self.warmup_factor = (
            1.0 / self.warmup_updates if self.warmup_updates > 0 else 1
        )
        self.end_learning_rate, self.power = args.end_learning_rate, args.power
        self.optimizer.set_lr(self.lr * self.warmup_factor)

    def reinit(self, total_num_update, num_updates):
        """Re-derive the ratio-based horizons after a checkpoint resume."""
        if self.warmup_ratio > 0:
            self.total_num_update = total_num_update
            self.warmup_updates = int(self.warmup_ratio * total_num_update)
        if 0 <= num_updates < self.warmup_updates:
            self.warmup_factor = (num_updates + 1) / float(self.warmup_updates)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--force-anneal", "--fa", type=int, metavar="N",
                            help="anneal from this epoch onward")
        parser.add_argument("--warmup-updates", default=0, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")
        parser.add_argument("--warmup-ratio", default=-1.0, type=float,
                            metavar="N",
                            help="warmup length as a fraction of total steps")
        parser.add_argument("--end-learning-rate", default=0.0, type=float,
                            metavar="LR")
        parser.add_argument("--power", default=1.0, type=float, metavar="P")
        parser.add_argument("--total-num-update", default=1000000, type=int,
                            metavar="N")

    def get_next_lr(self, epoch):
        schedule = self.args.lr
        anneal_from = self.args.force_anneal
        if anneal_from is not None and epoch >= anneal_from:
            # past force-anneal: hold whatever the update path set
            return self.optimizer.get_lr()
        return schedule[min(epoch, len(schedule) - 1)]

    def step_begin_epoch(self, epoch):
        self.lr = self.get_next_lr(epoch)
        self.optimizer.set_lr(self.lr * self.warmup_factor)
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        warmup, horizon = self.warmup_updates, self.total_num_update
        if warmup > 0 and num_updates <= warmup:
            self.warmup_factor = num_updates / float(warmup)
            lr = self.lr * self.warmup_factor
        elif num_updates >= horizon:
            lr = self.end_learning_rate
        else:
            remaining = 1 - (num_updates - warmup) / (horizon - warmup)
            span = self.lr - self.end_learning_rate
            lr = span * remaining**self.power + self.end_learning_rate
        self.optimizer.set_lr(lr)
        return self.optimizer.get_lr()
