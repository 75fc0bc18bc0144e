"""Masked-LM inference demo: load a checkpoint saved by `unicore-train`,
run batched no-grad prediction, report tokens/s.

Under `torch.no_grad()` the attention path switches to the flash kernel
automatically (linear memory in sequence length), so the same model that
trained at L=512 serves long contexts without the LxL score matrix.

Usage:
    python examples/bert/infer_demo.py --checkpoint ck.pt [--bf16] [--cpu]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)

from unicore_amd import checkpoint_utils, options, tasks


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--seq-len", type=int, default=512)
    p.add_argument("--iters", type=int, default=10)
    p.add_argument("--bf16", action="store_true")
    p.add_argument("--cpu", action="store_true")
    p.add_argument("--hip-graph", action="store_true",
                   help="capture the whole forward as a hipGraph and replay "
                        "it per batch (static shapes; removes all launch "
                        "overhead from the serving loop)")
    a = p.parse_args()

    state = checkpoint_utils.load_checkpoint_to_cpu(a.checkpoint)
    args = state["args"]
    args.cpu = a.cpu
    task = tasks.setup_task(args)
    model = task.build_model(args)
    model.load_state_dict(state["model"], strict=True, model_args=args)
    model.eval()
    use_cuda = torch.cuda.is_available() and not a.cpu
    if a.bf16:
        model = model.bfloat16()
    if use_cuda:
        model = model.cuda()

    vocab = len(task.dictionary)
    toks = torch.randint(5, vocab - 1, (a.batch_size, a.seq_len))
    mask_idx = getattr(task, "mask_idx", 4)
    toks[:, :: 7] = mask_idx  # mask every 7th position
    if use_cuda:
        toks = toks.cuda()

    def fwd(tokens):
        out = model(tokens)
        return out[0] if isinstance(out, tuple) else out

    with torch.no_grad():
        if a.hip_graph and use_cuda:
            # warm up twice on a side stream, then capture one replayable
            # graph over static in/out buffers
            static_in = toks.clone()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    fwd(static_in)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static_out = fwd(static_in)

            def run_batch(tokens):
                static_in.copy_(tokens)
                graph.replay()
                return static_out
        else:
            run_batch = fwd

        logits = run_batch(toks)
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(a.iters):
            logits = run_batch(toks)
        if use_cuda:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0

    pred = logits.float().argmax(-1)
    tps = a.batch_size * a.seq_len * a.iters / dt
    print(f"predicted shape {tuple(pred.shape)}; {tps:,.0f} tokens/s "
          f"({dt / a.iters * 1000:.1f} ms/batch)")


if __name__ == "__main__":
    main()
