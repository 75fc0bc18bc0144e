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
        parser.add_argument("--adadelta-rho", type=float, default=0.9,
                            metavar="RHO",
                            help="running-average coefficient for squared grads")
        parser.add_argument("--adadelta-eps", type=float, default=1e-6,
                            metavar="EPS",
                            help="denominator stability epsilon")
        parser.add_argument("--weight-decay", "--wd", default=0.0, type=float,
                            metavar="WD", help="weight decay")

    @property
    def optimizer_config(self):
        lr = self.args.lr
        return dict(
            lr=lr[0] if isinstance(lr, list) else lr,
            rho=self.args.adadelta_rho,
            eps=self.args.adadelta_eps,
            weight_decay=self.args.weight_decay,
        )

    @property
    def supports_flat_params(self):
        return True
