"""DDP engine factory (parity: reference
unicore/models/distributed_unicore_model.py:20-67).

``--ddp-backend`` maps onto our own engines (no dependency on torch c10d
DDP):

* ``c10d`` / ``pytorch_ddp`` / ``flat`` -> :class:`FlatDDP`: flat-bucket
  all-reduce on a side HIP stream, overlapped with backward (the MI355X
  replacement for torch DDP's bucketing).
* ``no_c10d`` / ``legacy_ddp`` -> :class:`LegacyDDP`: post-backward manual
  bucketing; the engine --allreduce-fp32-grad and --per-sample-clip-norm
  require.

The returned module is wrapped in :class:`ModuleProxyWrapper` so attribute
access and checkpoint (de)serialization reach the inner model.
"""

import logging

import torch.nn as nn

from unicore_amd.distributed import FlatDDP, LegacyDDP, ModuleProxyWrapper

logger = logging.getLogger(__name__)

_FLAT_NAMES = {"c10d", "pytorch_ddp", "flat"}
_LEGACY_NAMES = {"no_c10d", "legacy_ddp"}


def DistributedUnicoreModel(args, model, process_group, device):
    """Wrap *model* in the engine selected by ``args.ddp_backend`` and move
    it to *device*."""
    assert isinstance(model, nn.Module)
    if args.ddp_backend in _FLAT_NAMES:
        engine = FlatDDP(
            module=model.to(device),
            process_group=process_group,
            bucket_cap_mb=args.bucket_cap_mb,
        )
    elif args.ddp_backend in _LEGACY_NAMES:
        engine = LegacyDDP(
            module=model.to(device),
            process_group=process_group,
            buffer_size=2**28,
        )
    else:
        raise ValueError("Unknown --ddp-backend: " + args.ddp_backend)
    return ModuleProxyWrapper(engine)
