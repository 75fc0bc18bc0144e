"""DDP wrapper factory (parity: reference
unicore/models/distributed_unicore_model.py:20-67).

``--ddp-backend`` choices map to our own engines (no torch-c10d dependency):
  - ``c10d`` / ``pytorch_ddp``  -> FlatDDP (bucketed all-reduce on a side HIP
    stream overlapped with backward)
  - ``no_c10d`` / ``legacy_ddp`` -> LegacyDDP (post-backward bucketing;
    required for --allreduce-fp32-grad and --per-sample-clip-norm)
"""

import logging

from unicore_amd.distributed import FlatDDP, LegacyDDP, ModuleProxyWrapper

logger = logging.getLogger(__name__)


def DistributedUnicoreModel(args, model, process_group, device):
    """
    Wrap a *model* for distributed data-parallel training.

    This is similar to the legacy DistributedDataParallel module factory of
    the reference, picking the engine by ``--ddp-backend``.

    Args:
        args: command-line arguments
        model (BaseUnicoreModel): model to wrap
        process_group: the c10d process group to be used for distributed data
            parallel all-reduction.
        device: device to move model to
    """
    assert isinstance(model, __import__("torch").nn.Module)
    if args.ddp_backend in {"c10d", "pytorch_ddp", "flat"}:
        wrapped_model = FlatDDP(
            module=model.to(device),
            process_group=process_group,
            bucket_cap_mb=args.bucket_cap_mb,
        )
        # forward missing getattr and state_dict/load_state_dict to orig model
        wrapped_model = ModuleProxyWrapper(wrapped_model)
    elif args.ddp_backend in {"no_c10d", "legacy_ddp"}:
        wrapped_model = LegacyDDP(
            module=model.to(device),
            process_group=process_group,
            buffer_size=2**28,
        )
        wrapped_model = ModuleProxyWrapper(wrapped_model)
    else:
        raise ValueError("Unknown --ddp-backend: " + args.ddp_backend)

    return wrapped_model
