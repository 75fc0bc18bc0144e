"""Optimizer wrapper base (parity: reference
unicore/optim/unicore_optimizer.py:10-189).

Wraps a ``torch.optim.Optimizer`` and adds the trainer-facing surface:
grad scaling/clipping, the all-reduce hook, capability flags (queried off
the inner optimizer through one ``_inner_flag`` helper here), and
state-dict handling with override support.
"""

import torch

from unicore_amd import utils


class UnicoreOptimizer:
    def __init__(self, args):
        super().__init__()
        self.args = args

    @classmethod
    def add_args(cls, parser):
        """Hook for optimizer-specific CLI arguments."""

    @property
    def optimizer(self):
        """The wrapped torch.optim.Optimizer."""
        self._check_inner()
        return self._optimizer

    @optimizer.setter
    def optimizer(self, optimizer):
        """Swap the wrapped optimizer instance."""
        self._check_inner()
        self._optimizer = optimizer

    def _check_inner(self):
        inner = getattr(self, "_optimizer", None)
        if inner is None:
            raise NotImplementedError("subclass must create self._optimizer")
        if not isinstance(inner, torch.optim.Optimizer):
            raise ValueError("_optimizer must wrap a torch.optim.Optimizer")

    @property
    def optimizer_config(self):
        """kwargs that override optimizer state restored from checkpoints
        (so resumes can pick up e.g. a new learning rate)."""
        raise NotImplementedError("subclass must define optimizer_config")

    @property
    def params(self):
        """All parameters across all parameter groups."""
        for group in self.param_groups:
            yield from group["params"]

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    def __getstate__(self):
        return self._optimizer.__getstate__()

    def get_lr(self):
        """Learning rate of the first parameter group."""
        return self.param_groups[0]["lr"]

    def set_lr(self, lr) -> None:
        """Apply one learning rate to every parameter group."""
        for group in self.param_groups:
            group["lr"] = lr

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, state_dict, optimizer_overrides=None) -> None:
        """Restore optimizer state; *optimizer_overrides* (e.g. the lr from
        the current run's config) wins over the checkpointed values."""
        self.optimizer.load_state_dict(state_dict)
        if optimizer_overrides:
            for group in self.param_groups:
                group.update(optimizer_overrides)

    def backward(self, loss) -> None:
        """Backprop entry point (the fp16 wrapper scales here)."""
        loss.backward()

    def all_reduce_grads(self, module) -> None:
        """Trigger the module's manual grad all-reduce, when it has one."""
        hook = getattr(module, "all_reduce_grads", None)
        if hook is not None:
            hook()

    def multiply_grads(self, c) -> None:
        """Scale all existing gradients by *c* in a single foreach launch."""
        grads = [p.grad.data for p in self.params if p.grad is not None]
        if grads:
            if torch.is_tensor(c):
                c = c.to(grads[0].device)
            torch._foreach_mul_(grads, c)

    def clip_grad_norm(self, max_norm, aggregate_norm_fn=None):
        """Global L2-norm gradient clipping (multi-tensor kernel path)."""
        return utils.clip_grad_norm_(self.params, max_norm, aggregate_norm_fn)

    def per_sample_clip_grad_norm(self, max_norm, aggregate_norm_fn=None):
        raise NotImplementedError(
            "per-sample clipping requires the fp16/bf16 optimizer "
            "(--fp16 or --bf16 with --per-sample-clip-norm)"
        )

    def step(self, closure=None, scale=1.0, groups=None) -> None:
        """One optimization step; routes *scale* into the kernel when the
        inner optimizer fuses unscaling, otherwise pre-divides the grads."""
        if self.supports_step_with_scale:
            if self.supports_groups:
                self.optimizer.step(closure, scale=scale, groups=groups)
            else:
                self.optimizer.step(closure, scale=scale)
            return
        if scale != 1.0:
            self.multiply_grads(1.0 / scale)
        if self.supports_groups:
            self.optimizer.step(closure, groups=groups)
        else:
            self.optimizer.step(closure)

    def zero_grad(self) -> None:
        for param in self.params:
            param.grad = None
        self.optimizer.zero_grad()

    # capability flags, forwarded from the wrapped optimizer ---------------

    def _inner_flag(self, name):
        return getattr(self.optimizer, name, False)

    @property
    def supports_memory_efficient_fp16(self):
        return self._inner_flag("supports_memory_efficient_fp16")

    @property
    def supports_step_with_scale(self):
        return self._inner_flag("supports_step_with_scale")

    @property
    def supports_groups(self):
        return self._inner_flag("supports_groups")

    @property
    def supports_flat_params(self):
        """True when params/grads may be collapsed into one contiguous
        tensor per group."""
        return self._inner_flag("supports_flat_params")

    def broadcast_global_state_dict(self, state_dict):
        """For optimizers that shard state across ranks."""
        fn = getattr(self.optimizer, "broadcast_global_state_dict", None)
        return fn(state_dict) if fn is not None else state_dict
