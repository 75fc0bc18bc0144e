"""Module-level CPU tests: eager softmax_dropout (the numerics oracle),
broadcast descriptor logic, transformer encoder/decoder shapes,
LayerNorm/RMSNorm eager paths, metrics/meters.
"""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from unicore_amd.logging import meters, metrics
from unicore_amd.modules import (
    LayerNorm,
    RMSNorm,
    SelfMultiheadAttention,
    TransformerEncoder,
    softmax_dropout,
)
from unicore_amd.modules.softmax_dropout import _broadcast_descr


def test_broadcast_descr():
    # full match, no broadcast
    assert _broadcast_descr((4, 8), (4, 8)) == (32, 1)
    # leading-1 broadcast
    assert _broadcast_descr((1, 8), (4, 8)) == (8, 1)
    # trailing block broadcast: (4, 1) over (4, 8) is NOT a contiguous block
    assert _broadcast_descr((4, 1), (4, 8)) is None or _broadcast_descr((4, 1), (4, 8)) == (4, 8)
    # scalar
    assert _broadcast_descr((), (4, 8)) == (1, 1)
    # inner contiguous block
    descr = _broadcast_descr((8,), (4, 8))
    assert descr == (8, 1)


def test_softmax_dropout_eager_matches_reference_math():
    x = torch.randn(2, 3, 4, 5)
    mask = torch.randn(2, 1, 1, 5)
    bias = torch.randn(1, 3, 4, 5)
    out = softmax_dropout(x, 0.0, is_training=False, mask=mask, bias=bias)
    ref = F.softmax(x + mask + bias, dim=-1)
    assert torch.allclose(out, ref, atol=1e-6)


def test_softmax_dropout_training_cpu_drops():
    torch.manual_seed(0)
    x = torch.randn(64, 8, 16)
    out = softmax_dropout(x, 0.5, is_training=True)
    assert (out == 0).float().mean() > 0.2


def test_layernorm_rmsnorm_eager():
    ln = LayerNorm(32)
    x = torch.randn(4, 32)
    assert torch.allclose(ln(x), F.layer_norm(x, (32,), ln.weight, ln.bias, ln.eps))
    rn = RMSNorm(32)
    y = rn(x)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + rn.eps)
    assert torch.allclose(y, ref, atol=1e-5)


def test_transformer_encoder_forward():
    enc = TransformerEncoder(
        encoder_layers=2,
        embed_dim=64,
        ffn_embed_dim=128,
        attention_heads=4,
        emb_dropout=0.0,
        dropout=0.0,
        attention_dropout=0.0,
        activation_dropout=0.0,
        max_seq_len=32,
    )
    x = torch.randn(3, 16, 64)
    mask = torch.zeros(3, 16)
    out = enc(x, padding_mask=mask)
    assert out.shape == (3, 16, 64)


def test_self_attention_bias_path():
    attn = SelfMultiheadAttention(64, 4, dropout=0.0)
    x = torch.randn(2, 16, 64)  # (batch, seq, dim)
    bias = torch.zeros(2, 4, 16, 16)
    out = attn(x, attn_bias=bias)
    out2 = attn(x)
    assert out.shape == x.shape
    assert torch.allclose(out, out2, atol=1e-6)


def test_metrics_aggregation():
    with metrics.aggregate(new_root=True) as agg:
        metrics.log_scalar("loss", 2.0, weight=1)
        metrics.log_scalar("loss", 4.0, weight=3)
        vals = agg.get_smoothed_values()
    assert vals["loss"] == pytest.approx(3.5)


def test_metrics_nested_contexts():
    with metrics.aggregate(new_root=True) as outer:
        metrics.log_scalar("x", 1.0)
        with metrics.aggregate() as inner:
            metrics.log_scalar("x", 3.0)
        iv = inner.get_smoothed_values()
        ov = outer.get_smoothed_values()
    assert iv["x"] == pytest.approx(3.0)
    assert ov["x"] == pytest.approx(2.0)


def test_meters_state_dict_roundtrip():
    m = meters.AverageMeter()
    m.update(1.0)
    m.update(3.0)
    st = m.state_dict()
    m2 = meters.AverageMeter()
    m2.load_state_dict(st)
    assert m2.avg == pytest.approx(2.0)


def test_stopwatch_and_time_meter():
    t = meters.TimeMeter()
    t.update(5)
    assert t.n == 5
    sw = meters.StopwatchMeter()
    sw.start()
    sw.stop(n=2)
    assert sw.n == 2


def test_bert_post_ln_and_classification_head():
    """Model variants: post-LN trains a step on CPU; classification heads
    register and run (the fine-tuning surface of the reference BERT)."""
    import argparse

    import torch

    from unicore_amd.models.bert import BertModel, base_architecture

    args = argparse.Namespace()
    base_architecture(args)
    args.encoder_layers = 2
    args.encoder_embed_dim = 32
    args.encoder_ffn_embed_dim = 64
    args.encoder_attention_heads = 4
    args.max_seq_len = 64
    args.post_ln = True

    class D:
        def __len__(self):
            return 50

        def pad(self):
            return 1

    class T:
        dictionary = D()

    m = BertModel.build_model(args, T())
    toks = torch.randint(2, 49, (2, 16))
    logits = m(toks)
    out = logits[0] if isinstance(logits, tuple) else logits
    out.float().sum().backward()
    assert all(
        p.grad is None or torch.isfinite(p.grad.float()).all()
        for p in m.parameters()
    )

    m.register_classification_head("sent", num_classes=3)
    feats = m(toks, features_only=True, classification_head_name="sent")
    x = feats[0] if isinstance(feats, tuple) else feats
    assert x.shape[-1] == 3


def test_fused_op_eager_fallbacks_with_bias():
    """CPU eager fallbacks of the bias-carrying fused ops match explicit
    torch math (these paths also serve as the GPU numerics oracles)."""
    import torch

    from unicore_amd.modules import dropout_add, gated_mul, gelu_dropout

    torch.manual_seed(3)
    x = torch.randn(10, 16)
    res = torch.randn(10, 16)
    b = torch.randn(16)
    out = dropout_add(x, res, 0.0, True, bias=b)
    assert torch.allclose(out, res + x + b)

    g = torch.randn(10, 16)
    out = gelu_dropout(x, 0.0, True, bias=b)
    assert torch.allclose(out, torch.nn.functional.gelu(x + b))

    bg = torch.randn(16)
    out = gated_mul(x, g, b, bg)
    assert torch.allclose(out, (x + b) * torch.sigmoid(g + bg))
    out = gated_mul(x, g)
    assert torch.allclose(out, x * torch.sigmoid(g))


def test_msa_arrange_eager_fallback():
    import torch

    from unicore_amd.modules.msa_arrange import msa_arrange, msa_merge

    torch.manual_seed(4)
    B, S, L, H, D = 2, 3, 5, 4, 8
    x = torch.randn(B, S, L, H * D)
    for col in (False, True):
        a = msa_arrange(x, H, col)
        back = msa_merge(a, B, S, L, H, col)
        assert torch.allclose(back, x)


def test_broadcast_descr_matches_torch_semantics():
    """Property test: wherever _broadcast_descr claims a (src_nb, outer_div)
    addressing, the kernel's source-row formula must agree with torch's
    broadcast expansion for every batch row."""
    import itertools

    import torch

    from unicore_amd.modules.softmax_dropout import _broadcast_descr

    torch.manual_seed(9)
    q, k = 2, 4
    checked = 0
    for batch_dims in [(2,), (2, 3), (2, 3, 4)]:
        m = len(batch_dims)
        # every 0/1 pattern of which batch dims the source keeps
        for keep in itertools.product([False, True], repeat=m):
            src_batch = tuple(b if kp else 1 for b, kp in zip(batch_dims, keep))
            descr = _broadcast_descr(src_batch, list(batch_dims))
            if descr is None:
                continue
            src_nb, outer_div = descr
            src = torch.randn(*src_batch, q, k)
            full = src.expand(*batch_dims, q, k).reshape(-1, q, k)
            flat_src = src.reshape(-1, q, k)
            n_batch = full.shape[0]
            for b in range(n_batch):
                idx = (b // outer_div) % src_nb
                assert torch.equal(full[b], flat_src[idx]), (
                    src_batch, batch_dims, b, idx)
            checked += 1
    assert checked >= 10  # the contiguous-block patterns all verified


def test_qk_scores_matches_autograd_bmm():
    """qk_scores computes dk = dS^T @ q directly (contiguous layout) instead
    of differentiating through the k-transpose view; gradients must match
    plain bmm to fp64 precision, and dk must come out contiguous."""
    from unicore_amd.modules.multihead_attention import qk_scores

    torch.manual_seed(3)
    q = torch.randn(6, 10, 8, dtype=torch.float64, requires_grad=True)
    k = torch.randn(6, 10, 8, dtype=torch.float64, requires_grad=True)
    s = qk_scores(q, k)
    ds = torch.randn_like(s)
    s.backward(ds)
    dq_got, dk_got = q.grad.clone(), k.grad.clone()
    assert dk_got.is_contiguous()

    q2 = q.detach().clone().requires_grad_(True)
    k2 = k.detach().clone().requires_grad_(True)
    torch.bmm(q2, k2.transpose(1, 2)).backward(ds)
    assert torch.allclose(dq_got, q2.grad, atol=1e-12)
    assert torch.allclose(dk_got, k2.grad, atol=1e-12)

    # no-grad path returns the same values without the Function
    with torch.no_grad():
        assert torch.allclose(qk_scores(q.detach(), k.detach()), s.detach())


def test_dropout_add_ln_pre_eager_fallback():
    """CPU path of dropout_add_ln_pre: (summed, normed) must equal the
    manual compose, with gradients flowing through both outputs."""
    from unicore_amd.modules.dropout_add_ln import dropout_add_ln_pre

    torch.manual_seed(11)
    ln = torch.nn.LayerNorm(16)
    x = torch.randn(4, 6, 16, requires_grad=True)
    res = torch.randn(4, 6, 16, requires_grad=True)
    bias = torch.randn(16, requires_grad=True)

    s, n = dropout_add_ln_pre(x, res, ln, 0.0, True, bias=bias)
    (s.square().mean() + n.square().mean()).backward()
    got = (s.detach().clone(), n.detach().clone(),
           x.grad.clone(), res.grad.clone(), bias.grad.clone())

    for t in (x, res, bias):
        t.grad = None
    ln.weight.grad = ln.bias.grad = None
    s2 = res + x + bias
    n2 = ln(s2)
    (s2.square().mean() + n2.square().mean()).backward()
    assert torch.allclose(got[0], s2)
    assert torch.allclose(got[1], n2)
    assert torch.allclose(got[2], x.grad, atol=1e-6)
    assert torch.allclose(got[3], res.grad, atol=1e-6)
    assert torch.allclose(got[4], bias.grad, atol=1e-6)


def test_model_base_contract():
    """BaseUnicoreModel: set_num_updates must reach every submodule that
    defines it (without recursing into itself), and load_state_dict must
    accept the model_args kwarg (reference unicore/models/unicore_model.py
    :18-58 behavior contract)."""
    from unicore_amd.models.unicore_model import BaseUnicoreModel

    seen = []

    class Inner(torch.nn.Module):
        def set_num_updates(self, n):
            seen.append(n)

    class M(BaseUnicoreModel):
        def __init__(self):
            super().__init__()
            self.a = Inner()
            self.b = torch.nn.Sequential(Inner(), torch.nn.Linear(2, 2))

        def forward(self, x):
            return self.b[1](x)

    m = M()
    m.set_num_updates(7)
    assert seen == [7, 7]

    sd = m.state_dict()
    m2 = M()
    m2.load_state_dict(sd, strict=True, model_args=object())
    assert torch.equal(m2.b[1].weight, m.b[1].weight)
    # extract_features defaults to forward
    x = torch.randn(3, 2)
    assert torch.allclose(m.extract_features(x), m(x))
