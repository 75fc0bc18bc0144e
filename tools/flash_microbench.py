"""Op-level A/B: flash attention vs the materialized bmm+softmax chain at
BERT-base shapes (BH=1152, L=512, D=64, bias (H,L,L), dropout 0.1)."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    from unicore_amd import ops
    from unicore_amd.modules import softmax_dropout

    B, H, L, D = 96, 12, 512, 64
    BH = B * H
    p = 0.1
    q = torch.randn(BH, L, D, device="cuda", dtype=torch.bfloat16) * 0.2
    k, v = torch.randn_like(q), torch.randn_like(q)
    bias = torch.randn(H, L, L, device="cuda", dtype=torch.bfloat16)
    bias4 = bias.unsqueeze(0)

    # --- materialized forward chain -----------------------------------
    def mat_fwd():
        s = torch.bmm(q, k.transpose(1, 2)).view(B, H, L, L)
        attn = softmax_dropout(s, p, True, bias=bias4)
        return torch.bmm(attn.view(BH, L, L), v)

    # --- flash forward -------------------------------------------------
    def fl_fwd():
        return ops.flash_attn_fwd(q, k, v, bias, 1, None, 1, p, True)

    print(f"materialized fwd: {timeit(mat_fwd):8.2f} ms")
    print(f"flash fwd:        {timeit(fl_fwd):8.2f} ms")

    # --- full fwd+bwd through autograd ---------------------------------
    def mat_fb():
        qq = q.detach().requires_grad_(True)
        kk = k.detach().requires_grad_(True)
        vv = v.detach().requires_grad_(True)
        bb = bias4.detach().requires_grad_(True)
        s = torch.bmm(qq, kk.transpose(1, 2)).view(B, H, L, L)
        attn = softmax_dropout(s, p, True, bias=bb)
        o = torch.bmm(attn.view(BH, L, L), vv)
        o.backward(torch.ones_like(o))

    from unicore_amd.modules.multihead_attention import _FlashAttn

    def fl_fb():
        qq = q.detach().requires_grad_(True)
        kk = k.detach().requires_grad_(True)
        vv = v.detach().requires_grad_(True)
        bb = bias.detach().requires_grad_(True)
        o = _FlashAttn.apply(qq, kk, vv, bb, 1, None, 1, p, True)
        o.backward(torch.ones_like(o))

    print(f"materialized f+b: {timeit(mat_fb, iters=10):8.2f} ms")
    print(f"flash f+b:        {timeit(fl_fb, iters=10):8.2f} ms")

    # memory: flash never materializes the score matrix
    torch.cuda.reset_peak_memory_stats()
    mat_fb()
    torch.cuda.synchronize()
    print(f"materialized peak: {torch.cuda.max_memory_allocated()/2**30:.2f} GiB")
    torch.cuda.reset_peak_memory_stats()
    fl_fb()
    torch.cuda.synchronize()
    print(f"flash peak:        {torch.cuda.max_memory_allocated()/2**30:.2f} GiB")




def long_seq():
    """Long-sequence attention: flash vs materialized as L grows.  The
    materialized path's L x L scores explode quadratically; flash stays
    linear in memory — sequence lengths beyond the reference's reach."""
    from unicore_amd import ops
    from unicore_amd.modules import softmax_dropout

    H, D = 8, 64
    for L, B in ((1024, 16), (2048, 8), (4096, 4), (8192, 2), (16384, 1)):
        BH = B * H
        q = torch.randn(BH, L, D, device="cuda", dtype=torch.bfloat16) * 0.2
        k, v = torch.randn_like(q), torch.randn_like(q)

        def fl():
            return ops.flash_attn_fwd(q, k, v, None, 1, None, 1, 0.0, True)

        fl_ms = timeit(fl, iters=10, warmup=3)
        try:
            def mat():
                s = torch.bmm(q, k.transpose(1, 2))
                attn = softmax_dropout(s.view(B, H, L, L), 0.0, True)
                return torch.bmm(attn.view(BH, L, L), v)

            mat_ms = timeit(mat, iters=10, warmup=3)
            mat_s = f"{mat_ms:8.2f} ms"
        except torch.cuda.OutOfMemoryError:
            mat_s = "     OOM"
        torch.cuda.empty_cache()
        # fwd+bwd via the module-level autograd functions
        from unicore_amd.modules.multihead_attention import _FlashAttn

        def fl_fb():
            qq = q.detach().requires_grad_(True)
            kk = k.detach().requires_grad_(True)
            vv = v.detach().requires_grad_(True)
            o = _FlashAttn.apply(qq, kk, vv, None, 1, None, 1, 0.1, True)
            o.backward(torch.ones_like(o))

        fl_fb_ms = timeit(fl_fb, iters=5, warmup=2)
        try:
            def mat_fb():
                qq = q.detach().requires_grad_(True)
                kk = k.detach().requires_grad_(True)
                vv = v.detach().requires_grad_(True)
                s2 = torch.bmm(qq, kk.transpose(1, 2))
                attn = softmax_dropout(s2.view(B, H, L, L), 0.1, True)
                o = torch.bmm(attn.view(BH, L, L), vv)
                o.backward(torch.ones_like(o))

            mat_fb_ms = timeit(mat_fb, iters=5, warmup=2)
            mat_fb_s = f"{mat_fb_ms:8.2f} ms"
        except torch.cuda.OutOfMemoryError:
            mat_fb_s = "     OOM"
        torch.cuda.empty_cache()
        print(f"L={L:6d} B={B}: fwd flash {fl_ms:8.2f} / mat {mat_s} | "
              f"f+b flash {fl_fb_ms:8.2f} / mat {mat_fb_s}")


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--long-seq", action="store_true")
    a = ap.parse_args()
    if a.long_seq:
        long_seq()
    else:
        main()
