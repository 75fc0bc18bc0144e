"""Adam(W) optimizer: fused HIP kernel on GPU, pure-torch fallback on CPU.

Parity: reference unicore/optim/adam.py:22-204 (the torch fallback is also
the numerics oracle for the fused kernel tests).
"""

import logging
import math
from collections.abc import Collection

import torch
import torch.optim

from . import register_optimizer
from .unicore_optimizer import UnicoreOptimizer

logger = logging.getLogger(__name__)


@register_optimizer("adam")
class UnicoreAdam(UnicoreOptimizer):
    """Adam optimizer for unicore, with decoupled weight decay (AdamW).

    Picks the fused HIP kernel implementation when params live on the GPU
    and the extension is available; otherwise the pure-torch implementation
    below (identical numerics).
    """

    def __init__(self, args, params):
        super().__init__(args)
        params = list(params)
        first = None
        for p in params:
            candidates = p["params"] if isinstance(p, dict) else [p]
            for q in candidates:
                if torch.is_tensor(q):
                    first = q
                    break
            if first is not None:
                break
        use_fused = first is not None and first.is_cuda
        if use_fused:
            from unicore_amd import ops

            if ops.has_kernels():
                from .fused_adam import FusedAdam

                logger.info("using fused AdamW (gfx950 HIP kernel)")
                self._optimizer = FusedAdam(params, **self.optimizer_config)
                return
            elif not ops.allow_eager_on_gpu():
                ops.require_kernels()
        self._optimizer = Adam(params, **self.optimizer_config)

    @classmethod
    def add_args(cls, parser):
        """Add optimizer-specific arguments to the parser."""
        parser.add_argument(
            "--adam-betas",
            default="(0.9, 0.999)",
            metavar="B",
            help="betas for Adam optimizer",
        )
        parser.add_argument(
            "--adam-eps",
            type=float,
            default=1e-8,
            metavar="D",
            help="epsilon for Adam optimizer",
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
        """
        Return a kwarg dictionary that will be used to override optimizer
        args stored in checkpoints.
        """
        return {
            "lr": self.args.lr[0]
            if isinstance(self.args.lr, Collection)
            else self.args.lr,
            "betas": eval(self.args.adam_betas)
            if isinstance(self.args.adam_betas, str)
            else self.args.adam_betas,
            "eps": self.args.adam_eps,
            "weight_decay": self.args.weight_decay,
        }


class Adam(torch.optim.Optimizer):
    r"""Pure-torch AdamW (decoupled weight decay); fp32 state for low-precision
    params. Numerics oracle for the fused kernel."""

    def __init__(
        self,
        params,
        lr=1e-3,
        betas=(0.9, 0.999),
        eps=1e-8,
        weight_decay=0,
        amsgrad=False,
    ):
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, amsgrad=amsgrad
        )
        super(Adam, self).__init__(params, defaults)

    @property
    def supports_memory_efficient_fp16(self):
        return True

    @property
    def supports_flat_params(self):
        return True

    def step(self, closure=None):
        """Performs a single optimization step."""
        loss = None
        if closure is not None:
            loss = closure()

        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad.data
                if grad.dtype in {torch.float16, torch.bfloat16}:
                    grad = grad.float()
                if grad.is_sparse:
                    raise RuntimeError(
                        "Adam does not support sparse gradients, "
                        "please consider SparseAdam instead"
                    )
                amsgrad = group.get("amsgrad", False)

                p_data_fp32 = p.data
                if p.data.dtype in {torch.float16, torch.bfloat16}:
                    p_data_fp32 = p_data_fp32.float()

                state = self.state[p]

                # State initialization
                if len(state) == 0:
                    state["step"] = 0
                    # Exponential moving average of gradient values
                    state["exp_avg"] = torch.zeros_like(p_data_fp32)
                    # Exponential moving average of squared gradient values
                    state["exp_avg_sq"] = torch.zeros_like(p_data_fp32)
                    if amsgrad:
                        # Maintains max of all exp. moving avg. of sq. grad. values
                        state["max_exp_avg_sq"] = torch.zeros_like(p_data_fp32)
                else:
                    state["exp_avg"] = state["exp_avg"].to(p_data_fp32)
                    state["exp_avg_sq"] = state["exp_avg_sq"].to(p_data_fp32)
                    if amsgrad:
                        state["max_exp_avg_sq"] = state["max_exp_avg_sq"].to(
                            p_data_fp32
                        )

                exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
                if amsgrad:
                    max_exp_avg_sq = state["max_exp_avg_sq"]
                beta1, beta2 = group["betas"]

                state["step"] += 1

                # Decay the first and second moment running average coefficient
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                if amsgrad:
                    # Maintains the maximum of all 2nd moment running avg. till now
                    torch.max(max_exp_avg_sq, exp_avg_sq, out=max_exp_avg_sq)
                    # Use the max. for normalizing running avg. of gradient
                    denom = max_exp_avg_sq.sqrt().add_(group["eps"])
                else:
                    denom = exp_avg_sq.sqrt().add_(group["eps"])

                bias_correction1 = 1 - beta1 ** state["step"]
                bias_correction2 = 1 - beta2 ** state["step"]
                step_size = group["lr"] * math.sqrt(bias_correction2) / bias_correction1

                if group["weight_decay"] != 0:
                    p_data_fp32.add_(
                        p_data_fp32, alpha=-group["weight_decay"] * group["lr"]
                    )

                p_data_fp32.addcdiv_(exp_avg, denom, value=-step_size)

                if p.data.dtype in {torch.float16, torch.bfloat16}:
                    p.data.copy_(p_data_fp32)

        return loss
