"""Loss base class (parity: reference unicore/losses/unicore_loss.py:14-68)."""

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
                self.padding_idx = (
                    tgt_dict.pad() if tgt_dict is not None else -100
                )

    @classmethod
    def add_args(cls, parser):
        """Add loss-specific arguments to the parser."""
        pass

    @classmethod
    def build_loss(cls, args, task):
        """Construct a loss from command-line args."""
        # arguments in the __init__.
        init_args = {}
        for p in inspect.signature(cls).parameters.values():
            if (
                p.kind == p.POSITIONAL_ONLY
                or p.kind == p.VAR_POSITIONAL
                or p.kind == p.VAR_KEYWORD
            ):
                # we haven't implemented inference for these argument types,
                # but PRs welcome :)
                raise NotImplementedError("{} not supported".format(p.kind))

            assert p.kind in {p.POSITIONAL_OR_KEYWORD, p.KEYWORD_ONLY}

            if p.name == "task":
                init_args["task"] = task
            elif p.name == "args":
                init_args["args"] = args
            elif hasattr(args, p.name):
                init_args[p.name] = getattr(args, p.name)
            elif p.default != p.empty:
                pass  # we'll use the default value
            else:
                raise NotImplementedError(
                    "Unable to infer Loss arguments, please implement "
                    "{}.build_loss".format(cls.__name__)
                )
        return cls(**init_args)

    def forward(self, model, sample, reduce=True):
        """Compute the loss for the given sample.

        Returns a tuple with three elements:
        1) the loss
        2) the sample size, which is used as the denominator for the gradient
        3) logging outputs to display while training
        """
        raise NotImplementedError

    @staticmethod
    def reduce_metrics(logging_outputs: List[Dict[str, Any]], split="valid") -> None:
        """Aggregate logging outputs from data parallel training."""
        raise NotImplementedError

    @staticmethod
    def logging_outputs_can_be_summed(is_train: bool) -> bool:
        """
        Whether the logging outputs returned by `forward` can be summed
        across workers prior to calling `reduce_metrics`. Setting this
        to True will improve distributed training speed.
        """
        return False
