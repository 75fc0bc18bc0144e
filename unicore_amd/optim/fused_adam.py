"""FusedAdam: python wrapper over the gfx950 fused AdamW kernel.

Parity: reference unicore/optim/fused_adam.py:20-143. The kernel fuses grad
unscaling (``g / grad_scale``), both moment updates, bias correction and the
decoupled weight-decay parameter update into a single grid-stride pass per
flat tensor; m/v state is kept in fp32 while p/g may be fp16/bf16/fp32
(reference csrc/adam/adam_kernel.cu:16-46). With the flattened fp16/bf16
optimizer there are only ~2 flat tensors per step, so kernel-launch count is
negligible.
"""

import torch


class FusedAdam(torch.optim.Optimizer):
    """AdamW with fused HIP kernel.

    Compared to the eager path, ``step`` supports a fused grad scale:
    ``step(scale=s)`` divides grads by s inside the kernel without a separate
    sweep over memory.
    """

    def __init__(self, params, lr=1e-3, bias_correction=True,
                 betas=(0.9, 0.999), eps=1e-8, weight_decay=0.0,
                 amsgrad=False):
        if amsgrad:
            raise RuntimeError("FusedAdam does not support the AMSGrad variant.")
        super().__init__(params, dict(lr=lr, bias_correction=bias_correction,
                                      betas=betas, eps=eps,
                                      weight_decay=weight_decay))

    @property
    def supports_memory_efficient_fp16(self):
        return True

    @property
    def supports_flat_params(self):
        return True

    @property
    def supports_step_with_scale(self):
        return True

    def step(self, closure=None, scale=1.0):
        """One optimization step; ``scale`` divides the gradients inside
        the kernel (no separate unscale sweep over memory)."""
        from unicore_amd import ops

        loss = closure() if closure is not None else None
        for group in self.param_groups:
            bias_correction = group.get("bias_correction", True)
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                if p.grad.data.is_sparse:
                    raise RuntimeError("FusedAdam does not support sparse gradients")
                state = self.state[p]
                if len(state) == 0:
                    # moments live in fp32 regardless of param dtype
                    state = self.state[p] = dict(
                        step=0,
                        exp_avg=torch.zeros_like(p.data, dtype=torch.float32),
                        exp_avg_sq=torch.zeros_like(p.data, dtype=torch.float32),
                    )
                state["step"] += 1
                ops.fused_adam(
                    p.data, state["exp_avg"], state["exp_avg_sq"], p.grad.data,
                    group["lr"], beta1, beta2, group["eps"], scale,
                    state["step"], bias_correction, group["weight_decay"],
                )
        return loss
