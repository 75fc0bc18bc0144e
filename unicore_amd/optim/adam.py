"""Adam(W) optimizer: fused HIP kernel on GPU, pure-torch fallback on CPU.

Parity: reference unicore/optim/adam.py:22-204. The torch implementation is
decoupled-weight-decay AdamW with fp32 moments/master math for low-precision
params, and doubles as the numerics oracle for the fused kernel tests.
"""

import logging
import math
from collections.abc import Collection

import torch
import torch.optim

from . import register_optimizer
from .unicore_optimizer import UnicoreOptimizer

logger = logging.getLogger(__name__)


def _first_tensor(params):
    """The first actual tensor in a params list / param-group list."""
    for entry in params:
        for t in entry["params"] if isinstance(entry, dict) else [entry]:
            if torch.is_tensor(t):
                return t
    return None


@register_optimizer("adam")
class UnicoreAdam(UnicoreOptimizer):
    """AdamW front end: routes to the gfx950 fused kernel when the params
    are CUDA-resident and the extension loaded, else to the torch math."""

    def __init__(self, args, params):
        super().__init__(args)
        params = list(params)
        probe = _first_tensor(params)
        if probe is not None and probe.is_cuda:
            from unicore_amd import ops

            if ops.has_kernels():
                from .fused_adam import FusedAdam

                logger.info("using fused AdamW (gfx950 HIP kernel)")
                self._optimizer = FusedAdam(params, **self.optimizer_config)
                return
            if not ops.allow_eager_on_gpu():
                ops.require_kernels()
        self._optimizer = Adam(params, **self.optimizer_config)

    @classmethod
    def add_args(cls, parser):
        """Optimizer-specific CLI arguments."""
        parser.add_argument("--adam-betas", default="(0.9, 0.999)", metavar="B",
                            help="beta coefficients for Adam")
        parser.add_argument("--adam-eps", type=float, default=1e-8, metavar="D",
                            help="denominator epsilon for Adam")
        parser.add_argument("--weight-decay", "--wd", default=0.0, type=float,
                            metavar="WD", help="decoupled weight decay")

    @property
    def optimizer_config(self):
        """kwargs overriding checkpointed optimizer state on resume."""
        lr = self.args.lr
        betas = self.args.adam_betas
        return {
            "lr": lr[0] if isinstance(lr, Collection) else lr,
            "betas": eval(betas) if isinstance(betas, str) else betas,
            "eps": self.args.adam_eps,
            "weight_decay": self.args.weight_decay,
        }


class Adam(torch.optim.Optimizer):
    """Pure-torch AdamW (decoupled weight decay).

    Low-precision params are promoted to an fp32 working copy for the
    update and written back, matching the fused kernel's internal math.
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0, amsgrad=False):
        super().__init__(params, dict(lr=lr, betas=betas, eps=eps,
                                      weight_decay=weight_decay,
                                      amsgrad=amsgrad))

    @property
    def supports_memory_efficient_fp16(self):
        return True

    @property
    def supports_flat_params(self):
        return True

    def _moments(self, state, ref, amsgrad):
        """Fetch (creating or re-casting) the fp32 moment buffers."""
        keys = ["exp_avg", "exp_avg_sq"] + (["max_exp_avg_sq"] if amsgrad else [])
        if len(state) == 0:
            state["step"] = 0
            for k in keys:
                state[k] = torch.zeros_like(ref)
        else:
            for k in keys:
                state[k] = state[k].to(ref)
        return [state[k] for k in keys]

    def _apply_one(self, p, group):
        grad = p.grad.data
        if grad.is_sparse:
            raise RuntimeError(
                "Adam does not support sparse gradients, "
                "please consider SparseAdam instead"
            )
        low_precision = p.data.dtype in (torch.float16, torch.bfloat16)
        if grad.dtype in (torch.float16, torch.bfloat16):
            grad = grad.float()
        master = p.data.float() if low_precision else p.data

        amsgrad = group.get("amsgrad", False)
        state = self.state[p]
        moments = self._moments(state, master, amsgrad)
        m, v = moments[0], moments[1]
        beta1, beta2 = group["betas"]
        state["step"] += 1

        # moment EMAs
        m.mul_(beta1).add_(grad, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
        if amsgrad:
            vmax = moments[2]
            torch.max(vmax, v, out=vmax)  # normalize by the historical max
            denom = vmax.sqrt().add_(group["eps"])
        else:
            denom = v.sqrt().add_(group["eps"])

        bc1 = 1 - beta1 ** state["step"]
        bc2 = 1 - beta2 ** state["step"]
        step_size = group["lr"] * math.sqrt(bc2) / bc1

        if group["weight_decay"] != 0:
            # decoupled decay applied to the parameter, not the gradient
            master.add_(master, alpha=-group["weight_decay"] * group["lr"])
        master.addcdiv_(m, denom, value=-step_size)

        if low_precision:
            p.data.copy_(master)

    def step(self, closure=None):
        """One optimization step over every group."""
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    self._apply_one(p, group)
        return loss
