"""LR scheduler base (parity: reference
unicore/optim/lr_scheduler/unicore_lr_scheduler.py:12-50).

Three hooks drive every schedule: ``step_begin_epoch`` / ``step`` at epoch
boundaries (the latter also tracks the best validation loss) and
``step_update`` after every optimizer update (the hot path,
reference unicore/trainer.py:865-874).
"""

from argparse import Namespace

from unicore_amd.optim import UnicoreOptimizer


class UnicoreLRScheduler:
    def __init__(self, args: Namespace, optimizer, total_train_steps):
        super().__init__()
        if optimizer is not None:
            if not isinstance(optimizer, UnicoreOptimizer):
                raise ValueError(
                    "optimizer must be an instance of UnicoreOptimizer"
                )
        self.args, self.optimizer = args, optimizer
        self.total_train_steps = total_train_steps
        self.best = None  # best validation loss observed so far

    @classmethod
    def add_args(cls, parser):
        """Hook for schedule-specific CLI arguments."""

    def state_dict(self) -> dict:
        return dict(best=self.best)

    def load_state_dict(self, state: dict) -> None:
        self.best = state["best"]

    def step_begin_epoch(self, epoch):
        """Adjust the LR as a new epoch starts."""

    def step(self, epoch, val_loss=None):
        """Epoch-end hook; records the best validation loss seen."""
        if val_loss is None:
            return
        self.best = val_loss if self.best is None else min(self.best, val_loss)

    def step_update(self, num_updates):
        """Per-update hook; returns the (possibly updated) LR."""
        return self.optimizer.get_lr()  # constant unless overridden

    def reinit(self, total_num_update, num_updates):
        """Re-derive schedule constants after a resume (optional)."""
