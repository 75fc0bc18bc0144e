"""Flash attention kernel parity tests (MI355X, bf16, D=64)."""

import math

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs an MI355X"
)


def _ref_attn(q, k, v, bias=None, mask=None):
    s = q.float() @ k.float().transpose(-1, -2)
    if bias is not None:
        s = s + bias.float()
    if mask is not None:
        s = s + mask.float()
    p = F.softmax(s, dim=-1)
    return p @ v.float(), s


@requires_gpu
@pytest.mark.parametrize("L", [64, 128, 512])
def test_flash_fwd_plain(L):
    from unicore_amd import ops

    torch.manual_seed(0)
    BH = 8
    q = torch.randn(BH, L, 64, device="cuda", dtype=torch.bfloat16) * 0.2
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    o, lse, seed = ops.flash_attn_fwd(q, k, v, None, 1, None, 1, 0.0, True)
    ref, s = _ref_attn(q, k, v)
    assert (o.float() - ref).abs().max().item() < 2e-2, (o.float() - ref).abs().max()
    ref_lse = torch.logsumexp(s, dim=-1)
    assert (lse - ref_lse).abs().max().item() < 1e-3


@requires_gpu
def test_flash_fwd_bias_mask():
    from unicore_amd import ops

    torch.manual_seed(1)
    B, H, L = 3, 4, 128
    q = torch.randn(B * H, L, 64, device="cuda", dtype=torch.bfloat16) * 0.2
    k, v = torch.randn_like(q), torch.randn_like(q)
    # bias (1, H, L, L) broadcast over batch -> kernel form (H, L, L), od=1
    bias = torch.randn(H, L, L, device="cuda", dtype=torch.bfloat16)
    # mask (B, 1, 1, L) -> kernel form (B, 1, L), od=H
    mask = torch.zeros(B, 1, L, device="cuda", dtype=torch.bfloat16)
    mask[:, :, -17:] = float(torch.finfo(torch.float16).min)
    o, lse, seed = ops.flash_attn_fwd(q, k, v, bias, 1, mask, H, 0.0, True)

    bias4 = bias.unsqueeze(0)
    mask4 = mask.view(B, 1, 1, L)
    q4 = q.view(B, H, L, 64)
    ref, _ = _ref_attn(q4, k.view(B, H, L, 64), v.view(B, H, L, 64), bias4, mask4)
    diff = (o.view(B, H, L, 64).float() - ref).abs().max().item()
    assert diff < 2e-2, diff


@requires_gpu
def test_flash_fwd_dropout_stats_and_determinism():
    from unicore_amd import ops

    torch.manual_seed(2)
    BH, L, p = 8, 256, 0.3
    q = torch.randn(BH, L, 64, device="cuda", dtype=torch.bfloat16) * 0.2
    k, v = torch.randn_like(q), torch.randn_like(q)
    torch.manual_seed(7)
    o1, lse1, seed1 = ops.flash_attn_fwd(q, k, v, None, 1, None, 1, p, True)
    torch.manual_seed(7)
    o2, lse2, seed2 = ops.flash_attn_fwd(q, k, v, None, 1, None, 1, p, True)
    assert torch.equal(o1, o2) and int(seed1) == int(seed2)
    # mean of dropout(P)V stays close to PV (unbiased dropout)
    ref, _ = _ref_attn(q, k, v)
    rel = (o1.float().mean(0) - ref.mean(0)).abs().mean() / ref.abs().mean()
    assert rel.item() < 0.5  # loose: only 8 batches of dropout noise averaged
    assert not torch.equal(o1.float(), ref.bfloat16().float())
    # LSE unaffected by dropout (dropout is post-softmax)
    o3, lse3, _ = ops.flash_attn_fwd(q, k, v, None, 1, None, 1, 0.0, True)
    assert (lse1 - lse3).abs().max().item() < 1e-5
