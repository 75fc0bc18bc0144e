"""Loss base class (parity: reference unicore/losses/unicore_loss.py:14-68).

Contract: ``forward(model, sample) -> (loss, sample_size, logging_output)``
where sample_size is the gradient denominator;
``reduce_metrics`` aggregates the per-rank logging outputs; the
``logging_outputs_can_be_summed`` flag enables the fast all-reduce stat
sync path in the trainer.
"""

import inspect
from typing import Any, Dict, List

from torch.nn.modules.loss import _Loss


class UnicoreLoss(_Loss):
    def __init__(self, task):
        super().__init__()
        self.task = task
        if task is not None:
            self.args = task.args
            if hasattr(task, "target_dictionary"):
                tgt_dict = task.target_dictionary
                self.padding_idx = tgt_dict.pad() if tgt_dict is not None else -100

    @classmethod
    def add_args(cls, parser):
        """Hook for loss-specific CLI arguments."""

    @classmethod
    def build_loss(cls, args, task):
        """Instantiate the loss, wiring constructor parameters from the args
        namespace by name (``task``/``args`` are passed through directly)."""
        kwargs = {}
        for param in inspect.signature(cls).parameters.values():
            if param.kind in (param.POSITIONAL_ONLY, param.VAR_POSITIONAL,
                              param.VAR_KEYWORD):
                raise NotImplementedError(
                    f"{param.kind} not supported"
                )
            assert param.kind in {param.POSITIONAL_OR_KEYWORD, param.KEYWORD_ONLY}
            if param.name == "task":
                kwargs["task"] = task
            elif param.name == "args":
                kwargs["args"] = args
            elif hasattr(args, param.name):
                kwargs[param.name] = getattr(args, param.name)
            elif param.default is param.empty:
                raise NotImplementedError(
                    "Unable to infer Loss arguments, please implement "
                    f"{cls.__name__}.build_loss"
                )
            # else: the declared default applies
        return cls(**kwargs)

    def forward(self, model, sample, reduce=True):
        """Run the model on *sample* and return
        ``(loss, sample_size, logging_output)``."""
        raise NotImplementedError("losses implement forward")

    @staticmethod
    def reduce_metrics(logging_outputs: List[Dict[str, Any]],
                       split="valid") -> None:
        """Fold the collected per-step logging outputs into metrics."""
        raise NotImplementedError("losses implement reduce_metrics")

    @staticmethod
    def logging_outputs_can_be_summed(is_train: bool) -> bool:
        """True when forward()'s logging outputs are plain sums, enabling
        the cheap cross-rank all-reduce sync instead of pickled gather."""
        return False
