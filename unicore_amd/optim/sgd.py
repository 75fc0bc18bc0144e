"""SGD wrapper (parity: reference unicore/optim/sgd.py:13)."""

import torch.optim

from . import register_optimizer
from .unicore_optimizer import UnicoreOptimizer


@register_optimizer("sgd")
class SGD(UnicoreOptimizer):
    def __init__(self, args, params):
        super().__init__(args)
        self._optimizer = torch.optim.SGD(params, **self.optimizer_config)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--momentum", default=0.0, type=float, metavar="M",
                            help="momentum factor")
        parser.add_argument("--weight-decay", "--wd", default=0.0, type=float,
                            metavar="WD", help="weight decay")

    @property
    def optimizer_config(self):
        lr = self.args.lr
        return dict(
            lr=lr[0] if isinstance(lr, list) else lr,
            momentum=self.args.momentum,
            weight_decay=self.args.weight_decay,
        )

    @property
    def supports_flat_params(self):
        return True
