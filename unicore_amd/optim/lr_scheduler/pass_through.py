"""Pass-through scheduler (parity: reference
unicore/optim/lr_scheduler/pass_through.py:11-32): the optimizer carries
its own schedule and this shim simply forwards all hooks to it."""

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("pass_through")
class PassThroughScheduleSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)
        inner = getattr(optimizer, "lr_scheduler", None)
        assert inner is not None, (
            "Pass-through schedule can only be used with optimizers with "
            "their own schedulers"
        )

    def state_dict(self) -> dict:
        return self.optimizer.lr_scheduler.state_dict()

    def load_state_dict(self, state: dict) -> None:
        self.optimizer.lr_scheduler.load_state_dict(state)

    def step_begin_epoch(self, epoch):
        return self.optimizer.lr_scheduler.step_begin_epoch(epoch)

    def step_update(self, num_updates):
        return self.optimizer.lr_scheduler.step_update(num_updates)
