"""LegacyDDP — post-backward bucketed all-reduce engine.

Functional replacement for the reference's ``no_c10d``/legacy engine
(reference unicore/distributed/legacy_distributed_data_parallel.py:27-166):
no backward overlap; after backward the trainer calls
``all_reduce_grads()``, which walks the parameters in name order, packs
grads into a coalescing buffer (default 256 MB) and all-reduces it bucket
by bucket, pre-divided by the world size. This engine is required for
``--allreduce-fp32-grad`` and ``--per-sample-clip-norm``, where the fp32
flat grads are reduced by the optimizer instead.

Parameters carrying an ``expert`` attribute are never synchronized (same
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
        # no point allocating more than the model itself occupies
        self.buffer_size = min(
            buffer_size, sum(p.numel() for p in module.parameters())
        )
        self.buffer = None
        self.accumulate_grads = False

    @contextmanager
    def no_sync(self):
        """Suppress grad sync inside the block (grad accumulation)."""
        saved = self.accumulate_grads
        self.accumulate_grads = True
        yield
        self.accumulate_grads = saved

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def _reduce_bucket(self, params):
        """Coalesce one bucket of grads, all-reduce, scatter back."""
        scratch = self.buffer
        any_grad = False
        if len(params) == 1:
            # single (possibly oversized) param: reduce its grad in place
            p = params[0]
            if p.grad is not None:
                scratch = p.grad.data
                any_grad = True
            elif p.numel() <= self.buffer.numel():
                scratch = scratch[: p.numel()]
                scratch.zero_()
            else:
                scratch = torch.zeros_like(p)
        else:
            cursor = 0
            for p in params:
                n = p.numel()
                window = scratch[cursor: cursor + n]
                if p.grad is not None:
                    window.copy_(p.grad.data.view(-1))
                    any_grad = True
                else:
                    window.zero_()
                cursor += n

        if any_grad:
            scratch.div_(self.world_size)  # pre-divide: reduce stays a sum
        dist.all_reduce(scratch, group=self.process_group)

        # scatter the reduced values back into the param grads
        if len(params) == 1:
            p = params[0]
            if p.grad is not None:
                p.grad.data.copy_(scratch)
            else:
                p.grad = scratch.clone()
        else:
            cursor = 0
            for p in params:
                n = p.numel()
                window = scratch[cursor: cursor + n].view_as(p)
                if p.grad is not None:
                    p.grad.data.copy_(window)
                else:
                    p.grad = window.clone()
                cursor += n

    # kept under the reference's name: the fp16 optimizer's
    # --allreduce-fp32-grad path calls it with the flat fp32 grads
    all_reduce_params = _reduce_bucket

    def all_reduce_grads(self):
        """Synchronize gradients; call once after the last micro-batch's
        backward."""
        if self.accumulate_grads or self.world_size == 1:
            return

        if self.buffer is None:
            self.buffer = next(self.module.parameters()).new(self.buffer_size)

        pending = []
        used = 0
        for name, param in self.module.named_parameters():
            if not param.requires_grad:
                continue
            if param.grad is None:
                param.grad = torch.zeros_like(param)
            if hasattr(param, "expert"):
                continue  # expert-tagged params keep rank-local grads
            if param.grad.requires_grad:
                raise RuntimeError(
                    "DistributedDataParallel only works with gradients that "
                    "don't require grad"
                )
            n = param.numel()
            if n > self.buffer.numel():
                self._reduce_bucket([param])  # oversized: reduce alone
                continue
            if used + n > self.buffer.numel():
                self._reduce_bucket(pending)
                pending, used = [], 0
            pending.append(param)
            used += n

        if pending:
            self._reduce_bucket(pending)

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)
