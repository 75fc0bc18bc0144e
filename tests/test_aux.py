"""Auxiliary-subsystem tests (CPU): NanDetector forensics, EMA math,
BufferedIterator prefetch, finetune-from-model weight-only load,
checkpoint pruning policies, utils helpers."""

import os
import sys

import numpy as np
import pytest
import torch

from unicore_amd import utils
from unicore_amd.data.iterators import BufferedIterator
from unicore_amd.ema import ExponentialMovingAverageModel
from unicore_amd.nan_detector import NanDetector


def test_nan_detector_catches_bad_module():
    model = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 4))

    def poison(module, inp, out):
        return out * float("nan")

    model[0].register_forward_hook(poison)
    with NanDetector(model):
        out = model(torch.randn(2, 4))
    # detector logs and does not crash; output really is NaN
    assert torch.isnan(out).any()


def test_ema_update_math():
    import argparse

    model = torch.nn.Linear(4, 4)
    args = argparse.Namespace(ema_decay=0.5)
    ema = ExponentialMovingAverageModel(args, model, decay=0.5)
    orig = [p.detach().clone() for p in model.parameters()]
    with torch.no_grad():
        for p in model.parameters():
            p.add_(1.0)
    ema.update(list(model.named_parameters()))
    for o, e in zip(orig, ema.model_ema.parameters()):
        assert torch.allclose(e, o * 0.5 + (o + 1.0) * 0.5, atol=1e-6)


def test_buffered_iterator_order_and_len():
    data = list(range(100))
    it = BufferedIterator(10, iter_with_len(data))
    out = list(it)
    assert out == data


class iter_with_len:
    def __init__(self, data):
        self.data = data

    def __iter__(self):
        return iter(self.data)

    def __len__(self):
        return len(self.data)


def test_clip_grad_norm_helper():
    g = [torch.ones(10, requires_grad=False) for _ in range(3)]
    params = []
    for t in g:
        p = torch.nn.Parameter(t.clone())
        p.grad = torch.ones_like(p)
        params.append(p)
    total = utils.clip_grad_norm_(params, max_norm=1.0)
    expected = (30.0) ** 0.5
    assert float(total) == pytest.approx(expected, rel=1e-5)
    new_norm = torch.sqrt(sum((p.grad**2).sum() for p in params))
    assert float(new_norm) == pytest.approx(1.0, rel=1e-4)


def test_fp32_to_bf16_sr_cpu_fallback():
    src = torch.randn(1000) * 0.01
    dst = torch.empty(1000, dtype=torch.bfloat16)
    utils.fp32_to_bf16_sr(src, dst)
    assert ((dst.float() - src).abs() <= src.abs() * 2**-7 + 1e-30).all()


def test_tensor_tree_helpers():
    tree = {"a": torch.ones(2), "b": [torch.zeros(3), torch.ones(1)]}
    out = utils.tensor_tree_map(lambda t: t + 1, tree)
    assert torch.equal(out["a"], torch.full((2,), 2.0))
    assert torch.equal(out["b"][0], torch.ones(3))

    x = torch.randn(2, 3, 4, 5)
    y = utils.permute_final_dims(x, (1, 0))
    assert y.shape == (2, 3, 5, 4)

    m = torch.tensor([[1.0, 0.0], [1.0, 1.0]])
    v = torch.tensor([[2.0, 100.0], [3.0, 5.0]])
    mm = utils.masked_mean(m, v, dim=-1)
    assert torch.allclose(mm, torch.tensor([2.0, 4.0]))


def test_checkpoint_sequential_matches_direct():
    torch.manual_seed(0)
    layers = [torch.nn.Linear(8, 8) for _ in range(4)]

    def blocks(x):
        for l in layers:
            x = torch.relu(l(x))
        return x

    x = torch.randn(4, 8, requires_grad=True)
    ref = blocks(x)
    ref.sum().backward()
    g_ref = x.grad.clone()
    x.grad = None

    fns = [
        (lambda l: (lambda t: torch.relu(l(t))))(l) for l in layers
    ]
    out = utils.checkpoint_sequential(fns, x)[0]
    assert torch.allclose(out, ref, atol=1e-6)
    out.sum().backward()
    assert torch.allclose(x.grad, g_ref, atol=1e-6)


def test_finetune_from_model(tmp_path, monkeypatch):
    """--finetune-from-model loads weights only and restarts training."""
    from unicore_cli import train as train_cli

    base = [
        "--task", "bert_synthetic", "--arch", "bert_base",
        "--loss", "masked_lm", "--optimizer", "adam",
        "--lr-scheduler", "fixed", "--lr", "1e-4",
        "--batch-size", "4", "--dataset-size", "16",
        "--tokens-per-sample", "32", "--vocab-size", "64",
        "--encoder-layers", "1", "--encoder-embed-dim", "32",
        "--encoder-ffn-embed-dim", "64", "--encoder-attention-heads", "2",
        "--log-format", "none", "--cpu", "--num-workers", "0",
    ]
    d1 = str(tmp_path / "run1")
    monkeypatch.setattr(sys, "argv", ["t"] + base + [
        "--save-dir", d1, "--max-update", "3"])
    train_cli.cli_main()
    ckpt = os.path.join(d1, "checkpoint_last.pt")
    assert os.path.exists(ckpt)

    d2 = str(tmp_path / "run2")
    monkeypatch.setattr(sys, "argv", ["t"] + base + [
        "--save-dir", d2, "--max-update", "2",
        "--finetune-from-model", ckpt])
    train_cli.cli_main()
    st = torch.load(os.path.join(d2, "checkpoint_last.pt"), weights_only=False)
    # finetune restarts update counting from 0
    assert st["optimizer_history"][-1]["num_updates"] == 2
