"""Exercise FlatDDP's CUDA side-stream path on a single GPU.

The gloo CPU tests cover the async-handle branch; this covers the branch
the 8-GPU run actually takes — all-reduce launched on a dedicated HIP
stream with hipEvent ordering against the compute stream — by building a
1-rank RCCL group and faking world_size=2 (the 1-rank all_reduce is a
no-op, so grads come out exactly halved by the pre-divide)."""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs an MI355X"
)


@requires_gpu
def test_flat_ddp_side_stream_path():
    from unicore_amd.distributed import FlatDDP

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29881")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        torch.manual_seed(0)
        m = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 32)
        ).cuda()
        ref = [p.detach().clone() for p in m.parameters()]

        ddp = FlatDDP(m, process_group=dist.group.WORLD, bucket_cap_mb=0.001)
        # force the CUDA comm-stream machinery (init skipped it at ws == 1)
        ddp.world_size = 2
        ddp._comm_stream = torch.cuda.Stream()

        x = torch.randn(16, 32, device="cuda")
        # reference grads from an identical single model
        m2 = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 32)
        ).cuda()
        m2.load_state_dict(m.state_dict())
        m2(x).pow(2).mean().backward()

        ddp(x).pow(2).mean().backward()
        ddp.finish_grad_sync()
        torch.cuda.synchronize()

        # 1-rank all_reduce is identity; pre-divide by the faked ws=2 halves
        for p, q in zip(m.parameters(), m2.parameters()):
            assert torch.allclose(p.grad, q.grad / 2, atol=1e-6), (
                (p.grad - q.grad / 2).abs().max()
            )

        # second backward after re-arming (prepare_for_backward path)
        ddp.zero_grad_buffers()
        ddp.prepare_for_backward() if hasattr(ddp, "prepare_for_backward") else None
        ddp(x).pow(2).mean().backward()
        ddp.finish_grad_sync()
        torch.cuda.synchronize()
        for p, q in zip(m.parameters(), m2.parameters()):
            assert torch.allclose(p.grad, q.grad / 2, atol=1e-6)
    finally:
        dist.destroy_process_group()
