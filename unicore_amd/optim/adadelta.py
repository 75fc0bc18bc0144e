"""Adadelta wrapper (parity: reference unicore/optim/adadelta.py:13)."""

import torch.optim

from . import register_optimizer
from .unicore_optimizer import UnicoreOptimizer


@register_optimizer("adadelta")
class Adadelta(UnicoreOptimizer):
    def __init__(self, args, params):
        super().__init__(args)
        self._optimizer = torch.optim.Adadelta(params, **self.optimizer_config)

    @classmethod
    def add_args(cls, parser):
        """Add optimizer-specific arguments to the parser."""
        parser.add_argument(
            "--adadelta-rho",
            type=float,
            default=0.9,
            metavar="RHO",
            help="coefficient used for computing a running average of squared gradients",
        )
        parser.add_argument(
            "--adadelta-eps",
            type=float,
            default=1e-6,
            metavar="EPS",
            help="term added to the denominator to improve numerical stability",
        )
        parser.add_argument(
            "--weight-decay",
            "--wd",
            default=0.0,
            type=float,
            metavar="WD",
            help="weight decay",
        )

    @property
    def optimizer_config(self):
        return {
            "lr": self.args.lr[0] if isinstance(self.args.lr, list) else self.args.lr,
            "rho": self.args.adadelta_rho,
            "eps": self.args.adadelta_eps,
            "weight_decay": self.args.weight_decay,
        }

    @property
    def supports_flat_params(self):
        return True
