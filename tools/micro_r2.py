#!/usr/bin/env python3
"""Round-2 micro A/Bs: softmax bias-major on/off, embedding backward vs
torch, on the BERT bench shapes. Run on a GPU box."""
import os
import sys

sys.path.insert(0, __file__.rsplit("/tools/", 1)[0] if "/tools/" in __file__ else ".")
import time

import torch


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def softmax_case():
    from unicore_amd.modules import softmax_dropout

    B, H, L = 127, 12, 512
    x = torch.randn(B, H, L, L, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(1, H, L, L, device="cuda", dtype=torch.bfloat16)

    def run():
        softmax_dropout(x, 0.1, True, bias=bias, inplace=False)

    print(f"softmax fwd (bias-major={os.environ.get('UNICORE_SM_BIASMAJOR', '1')}): "
          f"{timeit(run):.3f} ms")


def embedding_case():
    from unicore_amd import ops

    V, D, N = 30592, 768, 127 * 512
    grad = torch.randn(N, D, device="cuda", dtype=torch.bfloat16)
    idx = torch.randint(5, V, (N,), device="cuda")

    def fused():
        ops.embedding_bwd(grad, idx, V, 1)

    w = torch.zeros(V, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)

    def torch_path():
        out = torch.nn.functional.embedding(idx, w, padding_idx=1)
        out.backward(grad)
        w.grad = None

    print(f"embedding bwd fused : {timeit(fused):.3f} ms")
    print(f"embedding bwd torch (fwd+bwd): {timeit(torch_path):.3f} ms")


if __name__ == "__main__":
    softmax_case()
    embedding_case()
