"""LegacyDDP — post-backward bucketed all-reduce engine.

Functional replacement for the reference's ``no_c10d``/legacy engine
(reference unicore/distributed/legacy_distributed_data_parallel.py:27-166):
no backward overlap; after backward the trainer calls
``all_reduce_grads()`` which copies grads into a coalescing buffer
(default 256 MB) and all-reduces it bucket by bucket. Required for
``--allreduce-fp32-grad`` and ``--per-sample-clip-norm`` modes, where the
fp32 flat grads are reduced by the optimizer instead.

Parameters whose name ends with ``.expert`` are skipped during sync (same
escape hatch as the reference, legacy_distributed_data_parallel.py:142-144).
"""

import logging
from contextlib import contextmanager

import torch
import torch.distributed as dist
from torch import nn

logger = logging.getLogger(__name__)


class LegacyDDP(nn.Module):
    def __init__(self, module, process_group, buffer_size=2**28):
        super().__init__()
        self.module = module
        self.process_group = process_group
        self.world_size = (
            dist.get_world_size(process_group) if dist.is_initialized() else 1
        )
        # Never use a bigger buffer than the number of model params
        self.buffer_size = min(
            buffer_size, sum(p.numel() for p in module.parameters())
        )
        self.buffer = None
        self.accumulate_grads = False

        # We can also forcibly accumulate grads locally and only do the
        # all-reduce at some later time
        self._grad_sync_disabled = False

    @contextmanager
    def no_sync(self):
        """A context manager to disable gradient synchronization."""
        old = self.accumulate_grads
        self.accumulate_grads = True
        yield
        self.accumulate_grads = old

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def all_reduce_params(self, params):
        buffer = self.buffer
        nonzero_buffer = False
        if len(params) > 1:
            offset = 0
            for p in params:
                sz = p.numel()
                if p.grad is not None:
                    buffer[offset : offset + sz].copy_(p.grad.data.view(-1))
                    nonzero_buffer = True
                else:
                    buffer[offset : offset + sz].zero_()
                offset += sz
        else:
            # we only have a single grad to all-reduce
            p = params[0]
            if p.grad is not None:
                buffer = p.grad.data
                nonzero_buffer = True
            elif p.numel() <= self.buffer.numel():
                buffer = buffer[: p.numel()]
                buffer.zero_()
            else:
                buffer = torch.zeros_like(p)

        if nonzero_buffer:
            buffer.div_(self.world_size)

        dist.all_reduce(buffer, group=self.process_group)

        # copy all-reduced grads back into their original place
        if len(params) > 1:
            offset = 0
            for p in params:
                sz = p.numel()
                if p.grad is not None:
                    p.grad.data.copy_(buffer[offset : offset + sz].view_as(p))
                else:
                    p.grad = buffer[offset : offset + sz].view_as(p).clone()
                offset += sz
        else:
            p = params[0]
            if p.grad is not None:
                p.grad.data.copy_(buffer)
            else:
                p.grad = buffer.clone()

    def all_reduce_grads(self):
        """
        This function must be called explicitly after backward to reduce
        gradients.
        """
        if self.accumulate_grads or self.world_size == 1:
            return

        if self.buffer is None:
            first = next(self.module.parameters())
            self.buffer = first.new(self.buffer_size)

        buffered_params = []
        offset = 0
        for param_name, param in self.module.named_parameters():
            if not param.requires_grad:
                continue
            if param.grad is None:
                param.grad = torch.zeros_like(param)
            if hasattr(param, "expert"):
                # skip synchronizing grads for expert-tagged params
                continue
            if param.grad.requires_grad:
                raise RuntimeError(
                    "DistributedDataParallel only works with gradients that "
                    "don't require grad"
                )
            sz = param.numel()
            if sz > self.buffer.numel():
                # all-reduce big params directly
                self.all_reduce_params([param])
            else:
                if offset + sz > self.buffer.numel():
                    self.all_reduce_params(buffered_params)
                    offset = 0
                    buffered_params.clear()
                buffered_params.append(param)
                offset += sz

        if len(buffered_params) > 0:
            self.all_reduce_params(buffered_params)

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)
