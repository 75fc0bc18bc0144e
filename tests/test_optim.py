"""Optimizer / LR-scheduler / loss-scaler semantics tests (CPU).

These pin the behavioral contracts listed in SURVEY.md Appendix A and the
fp16/bf16 optimizer invariants of reference unicore/optim/fp16_optimizer.py.
"""

import argparse
import math

import pytest
import torch

from unicore_amd import optim
from unicore_amd.optim import lr_scheduler as lrs
from unicore_amd.optim.dynamic_loss_scaler import DynamicLossScaler


def _ns(**kw):
    ns = argparse.Namespace()
    for k, v in kw.items():
        setattr(ns, k, v)
    return ns


def _mk_optimizer(lr=0.1, **extra):
    model = torch.nn.Linear(4, 4)
    args = _ns(
        optimizer="adam",
        lr=[lr],
        adam_betas="(0.9, 0.999)",
        adam_eps=1e-8,
        weight_decay=0.0,
        **extra,
    )
    return args, optim.build_optimizer(args, list(model.named_parameters())), model


def test_adam_matches_torch_adamw():
    torch.manual_seed(0)
    x = torch.randn(32, 16)
    w1 = torch.nn.Linear(16, 16)
    w2 = torch.nn.Linear(16, 16)
    w2.load_state_dict(w1.state_dict())

    args = _ns(
        optimizer="adam",
        lr=[1e-2],
        adam_betas="(0.9, 0.999)",
        adam_eps=1e-8,
        weight_decay=0.0,
    )
    ours = optim.build_optimizer(args, list(w1.named_parameters()))
    theirs = torch.optim.AdamW(w2.parameters(), lr=1e-2, betas=(0.9, 0.999),
                               eps=1e-8, weight_decay=0.0)
    for _ in range(5):
        ours.zero_grad()
        w1(x).pow(2).mean().backward()
        ours.step()
        theirs.zero_grad()
        w2(x).pow(2).mean().backward()
        theirs.step()
    for p1, p2 in zip(w1.parameters(), w2.parameters()):
        # our Adam adds eps after sqrt of the UN-bias-corrected v (reference
        # semantics) -> tiny drift vs torch.AdamW is expected
        assert torch.allclose(p1, p2, atol=5e-5), (p1 - p2).abs().max()


def test_clip_grad_norm():
    _, opt, model = _mk_optimizer()
    model(torch.randn(8, 4)).sum().backward()
    total = opt.clip_grad_norm(0.001)
    g2 = torch.sqrt(sum((p.grad**2).sum() for p in model.parameters()))
    assert g2 <= 0.0011
    assert total > 0


def test_multiply_grads():
    _, opt, model = _mk_optimizer()
    model(torch.randn(8, 4)).sum().backward()
    before = [p.grad.clone() for p in model.parameters()]
    opt.multiply_grads(0.5)
    for b, p in zip(before, model.parameters()):
        assert torch.allclose(p.grad, b * 0.5)


def _sched(name, lr=1.0, total=1000, **kw):
    defaults = dict(
        lr_scheduler=name,
        lr=[lr],
        optimizer="adam",
        adam_betas="(0.9, 0.999)",
        adam_eps=1e-8,
        weight_decay=0.0,
        force_anneal=None,
        lr_shrink=0.1,
        warmup_updates=0,
    )
    defaults.update(kw)
    args = _ns(**defaults)
    model = torch.nn.Linear(2, 2)
    opt = optim.build_optimizer(args, list(model.named_parameters()))
    return lrs.build_lr_scheduler(args, opt, total), opt


def test_fixed_schedule():
    s, opt = _sched("fixed")
    assert s.step_update(10) == pytest.approx(1.0)


def test_polynomial_decay():
    s, opt = _sched(
        "polynomial_decay",
        warmup_updates=100,
        warmup_ratio=-1,
        end_learning_rate=0.0,
        power=1.0,
        total_num_update=1000,
    )
    assert s.step_update(50) == pytest.approx(0.5)
    assert s.step_update(100) == pytest.approx(1.0)
    assert s.step_update(550) == pytest.approx(0.5, rel=1e-3)
    assert s.step_update(1000) == pytest.approx(0.0, abs=1e-6)


def test_inverse_sqrt():
    s, opt = _sched("inverse_sqrt", warmup_updates=100, warmup_init_lr=-1)
    assert s.step_update(100) == pytest.approx(1.0)
    assert s.step_update(400) == pytest.approx(0.5)


def test_cosine_schedule():
    s, opt = _sched(
        "cosine",
        warmup_updates=10,
        warmup_ratio=-1,
        warmup_init_lr=-1,
        min_lr=0.0,
        max_lr=1.0,
        t_mult=1,
        lr_period_updates=90,
    )
    s.step_update(10)
    top = opt.get_lr()
    s.step_update(55)
    mid = opt.get_lr()
    assert top == pytest.approx(1.0, rel=1e-2)
    assert 0.4 < mid < 0.6


def test_exponential_decay():
    s, opt = _sched(
        "exponential_decay",
        warmup_updates=0,
        decay_ratio=0.5,
        decay_steps=100,
        stair_decay=False,
    )
    assert s.step_update(100) == pytest.approx(0.5)
    assert s.step_update(200) == pytest.approx(0.25)


def test_tri_stage():
    s, opt = _sched(
        "tri_stage",
        warmup_steps=100,
        hold_steps=100,
        decay_steps=100,
        init_lr_scale=0.01,
        final_lr_scale=0.05,
        phase_ratio=None,
        max_update=300,
    )
    assert s.step_update(0) == pytest.approx(0.01)
    assert s.step_update(100) == pytest.approx(1.0)
    assert s.step_update(150) == pytest.approx(1.0)
    assert s.step_update(300) == pytest.approx(0.05, rel=1e-2)


def test_pass_through_scheduler_requires_internal_schedule():
    # pass_through is only valid for optimizers that schedule their own LR
    with pytest.raises(AssertionError):
        _sched("pass_through")


def test_dynamic_loss_scaler_growth_and_backoff():
    s = DynamicLossScaler(init_scale=128.0, scale_window=4, tolerance=0.0)
    for _ in range(5):
        s.update()
    # one growth after a full no-overflow window
    assert s.loss_scale == 128.0 * 2
    with pytest.raises(OverflowError):
        s.check_overflow(float("inf"))
    assert s.loss_scale == 128.0  # halved on overflow
    with pytest.raises(FloatingPointError):
        tiny = DynamicLossScaler(init_scale=2e-5, scale_window=4,
                                 min_loss_scale=1e-4)
        tiny.check_overflow(float("nan"))


def test_fp16_flatten_roundtrip():
    # flattened bf16 params + fp32 master: step updates both consistently
    model = torch.nn.Sequential(torch.nn.Linear(8, 8), torch.nn.Linear(8, 8))
    model = model.bfloat16()
    args = _ns(
        optimizer="adam",
        lr=[1e-2],
        adam_betas="(0.9, 0.999)",
        adam_eps=1e-8,
        weight_decay=0.0,
        bf16=True,
        bf16_sr=False,
        fp16=False,
        allreduce_fp32_grad=False,
        fp16_no_flatten_grads=False,
        fp16_init_scale=4,
        fp16_scale_window=None,
        fp16_scale_tolerance=0.0,
        min_loss_scale=1e-4,
        threshold_loss_scale=None,
        distributed_world_size=1,
        update_freq=[1],
        no_weight_decay_names="",
        per_sample_clip_norm=0.0,
    )
    fopt = optim.FP16Optimizer.build_optimizer(args, list(model.named_parameters()))
    x = torch.randn(4, 8).bfloat16()
    loss = model(x).float().pow(2).mean()
    fopt.backward(loss)
    gnorm = fopt.clip_grad_norm(1.0)
    assert torch.isfinite(torch.as_tensor(float(gnorm)))
    before = [p.detach().clone() for p in model.parameters()]
    fopt.step()
    changed = any(
        not torch.equal(b, p.detach()) for b, p in zip(before, model.parameters())
    )
    assert changed
    # fp32 master mirrors the bf16 params after writeback
    for f in fopt.fp32_params:
        assert f.dtype == torch.float32
    # every bf16 param equals its fp32 master segment rounded to bf16
    for p in model.parameters():
        master = fopt.fp32_view_of(p)
        assert torch.equal(p.detach(), master.bfloat16())


def test_lazy_grad_collection_matches_view_path():
    """The single-process lazy path (autograd-assigned grads batch-copied
    into the fp32 masters) must match the flat-view path step for step,
    including grad accumulation and a parameter with no grad."""
    import argparse
    import copy

    import torch

    from unicore_amd.optim import FP16Optimizer

    def make(seed):
        torch.manual_seed(seed)
        m = torch.nn.Sequential(
            torch.nn.Linear(16, 32), torch.nn.Tanh(), torch.nn.Linear(32, 16),
            torch.nn.Linear(16, 4),  # untouched in step 2 (no grad)
        ).bfloat16()
        return m

    args = argparse.Namespace(
        optimizer="adam", lr=[1e-2], adam_betas="(0.9, 0.98)", adam_eps=1e-8,
        weight_decay=0.01, bf16=True, bf16_sr=False, fp16=False,
        allreduce_fp32_grad=False, fp16_no_flatten_grads=False,
        min_loss_scale=1e-4, fp16_scale_window=None, fp16_scale_tolerance=0.0,
        fp16_init_scale=4, threshold_loss_scale=None, per_sample_clip_norm=0.0,
        distributed_world_size=1, update_freq=[1],
    )

    ma, mb = make(7), make(7)
    oa = FP16Optimizer.build_optimizer(args, list(ma.named_parameters()))
    ob = FP16Optimizer.build_optimizer(args, list(mb.named_parameters()))
    ob.enable_lazy_grad_collection()

    x = torch.randn(8, 16).bfloat16()
    for step in range(3):
        for m, o in ((ma, oa), (mb, ob)):
            o.zero_grad()
            # grad accumulation: two backwards per step
            for micro in range(2):
                h = m[1](m[0](x + step * 0.1 + micro * 0.01))
                if step == 2:
                    loss = m[2](h).float().pow(2).mean()  # m[3] gets no grad
                else:
                    loss = m[3](m[2](h)).float().pow(2).mean()
                o.backward(loss)
            o.multiply_grads(0.5)
            o.clip_grad_norm(1.0)
            o.step()
        for (na, pa), (nb, pb) in zip(ma.named_parameters(), mb.named_parameters()):
            assert torch.equal(pa, pb), (step, na)
