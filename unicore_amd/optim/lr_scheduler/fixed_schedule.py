"""Fixed / force-annealed LR schedule (parity: reference
unicore/optim/lr_scheduler/fixed_schedule.py:13-69).

Per-epoch LR comes straight from the ``--lr`` list (last entry repeats);
after ``--force-anneal`` the last entry decays by ``lr_shrink`` per epoch.
A linear warmup factor ramps the applied LR over the first
``--warmup-updates`` steps.
"""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("fixed")
class FixedSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        self.lr = args.lr[0]
        self.warmup_factor = (
            1.0 / args.warmup_updates if args.warmup_updates > 0 else 1
        )

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--force-anneal", "--fa", type=int, metavar="N",
                            help="anneal from this epoch onward")
        parser.add_argument("--lr-shrink", default=0.1, type=float,
                            metavar="LS",
                            help="annealing shrink factor, lr_new = lr * lr_shrink")
        parser.add_argument("--warmup-updates", default=0, type=int,
                            metavar="N",
                            help="linear LR warmup over the first N updates")

    def state_dict(self) -> dict:
        return {"lr": self.lr}

    def load_state_dict(self, state: dict) -> None:
        if "lr" in state:
            self.lr = state["lr"]

    def get_next_lr(self, epoch):
        schedule = self.args.lr
        anneal_from = self.args.force_anneal
        if anneal_from is None or epoch < anneal_from:
            # inside the explicit schedule (last entry repeats)
            return schedule[min(epoch - 1, len(schedule) - 1)]
        return schedule[-1] * self.args.lr_shrink ** (epoch + 1 - anneal_from)

    def step_begin_epoch(self, epoch):
        self.lr = self.get_next_lr(epoch)
        self.optimizer.set_lr(self.warmup_factor * self.lr)
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        warmup = self.args.warmup_updates
        if warmup > 0 and num_updates < warmup:
            self.warmup_factor = (num_updates + 1) / float(warmup)
            self.optimizer.set_lr(self.warmup_factor * self.lr)
        else:
            self.optimizer.set_lr(self.lr)
        return self.optimizer.get_lr()
