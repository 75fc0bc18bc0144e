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


@requires_gpu
def test_flat_ddp_lazy_aliased_side_stream():
    """Aliased (bf16 optimizer flats) + lazy grad collection on the CUDA
    comm-stream path: 1-rank RCCL group with faked world_size=2, so the
    pre-divide halves the reference grads exactly."""
    import argparse

    from unicore_amd.distributed import FlatDDP
    from unicore_amd.optim import FP16Optimizer

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29882")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        torch.manual_seed(0)
        m = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 32)
        ).cuda().bfloat16()
        m2 = torch.nn.Sequential(
            torch.nn.Linear(32, 64), torch.nn.ReLU(), torch.nn.Linear(64, 32)
        ).cuda().bfloat16()
        m2.load_state_dict(m.state_dict())

        args = argparse.Namespace(
            optimizer="adam", lr=[1e-2], adam_betas="(0.9, 0.98)",
            adam_eps=1e-8, weight_decay=0.0, bf16=True, bf16_sr=False,
            fp16=False, allreduce_fp32_grad=False,
            fp16_no_flatten_grads=False, min_loss_scale=1e-4,
            fp16_scale_window=None, fp16_scale_tolerance=0.0,
            fp16_init_scale=4, threshold_loss_scale=None,
            per_sample_clip_norm=0.0, distributed_world_size=2,
            update_freq=[1],
        )
        opt = FP16Optimizer.build_optimizer(args, list(m.named_parameters()))
        ddp = FlatDDP(m, process_group=dist.group.WORLD, bucket_cap_mb=0.001)
        ddp.world_size = 2
        ddp.lazy = True
        ddp._comm_stream = torch.cuda.Stream()
        assert all(p.grad is None for p in m.parameters()) or ddp.lazy

        x = torch.randn(16, 32, device="cuda").bfloat16()
        m2(x).float().pow(2).mean().backward()

        for p in m.parameters():
            p.grad = None  # ddp constructed with ws=1 kept the views
        ddp(x).float().pow(2).mean().backward()
        ddp.finish_grad_sync()
        torch.cuda.synchronize()

        # grads landed in the bucket views (optimizer lp flats), halved
        # by the ws=2 pre-divide; p.grad was consumed by the bucket copy
        assert all(p.grad is None for p in m.parameters())
        ref = {id(p): q for p, q in zip(m.parameters(), m2.parameters())}
        for b in ddp._buckets:
            for p, v in zip(b.params, b.views):
                want = (ref[id(p)].grad.float() / 2).view_as(v)
                assert torch.allclose(v.float(), want, atol=2e-2), (
                    (v.float() - want).abs().max()
                )
    finally:
        dist.destroy_process_group()
