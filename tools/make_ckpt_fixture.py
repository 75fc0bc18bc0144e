#!/usr/bin/env python3
"""Generate tests/fixtures/reference_format_checkpoint.pt — a checkpoint in
the EXACT reference on-disk schema (reference
unicore/trainer.py:258-284 + unicore/checkpoint_utils.py:280-284), built
field by field so the test pins our reader against the documented layout,
not against whatever our writer happens to emit.

Run once and commit the artifact:
    python tools/make_ckpt_fixture.py
"""

import os
from argparse import Namespace

import torch

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
OUT = os.path.join(REPO, "tests", "fixtures",
                   "reference_format_checkpoint.pt")

# the tiny model config the loading test reconstructs
TINY = dict(
    arch="bert_base",
    encoder_layers=2,
    encoder_embed_dim=64,
    encoder_ffn_embed_dim=128,
    encoder_attention_heads=2,
    max_seq_len=66,
    vocab_size=128,
)


def tiny_model_state():
    import sys

    sys.path.insert(0, REPO)
    from unicore_amd import options, tasks

    argv = [
        "--task", "bert_synthetic",
        "--arch", TINY["arch"],
        "--loss", "masked_lm",
        "--optimizer", "adam",
        "--lr-scheduler", "fixed",
        "--lr", "1e-4",
        "--batch-size", "2",
        "--dataset-size", "8",
        "--tokens-per-sample", "64",
        "--max-seq-len", str(TINY["max_seq_len"]),
        "--vocab-size", str(TINY["vocab_size"]),
        "--encoder-layers", str(TINY["encoder_layers"]),
        "--encoder-embed-dim", str(TINY["encoder_embed_dim"]),
        "--encoder-ffn-embed-dim", str(TINY["encoder_ffn_embed_dim"]),
        "--encoder-attention-heads", str(TINY["encoder_attention_heads"]),
        "--seed", "11",
        "--cpu",
    ]
    parser = options.get_training_parser()
    args = options.parse_args_and_arch(parser, input_args=argv)
    torch.manual_seed(11)
    task = tasks.setup_task(args)
    model = task.build_model(args)
    return args, model.state_dict()


def main():
    args, model_state = tiny_model_state()

    # Appendix-B layout, written literally (reference unicore/trainer.py:258-284)
    state = {
        "args": Namespace(**vars(args)),
        "model": model_state,
        "loss": None,
        "optimizer_history": [
            {
                "loss_name": "MaskedLMLoss",
                "optimizer_name": "Adam",
                "lr_scheduler_state": {"lr": 1e-4},
                "num_updates": 7,
            }
        ],
        "task_state": {},
        "extra_state": {
            "metrics": {},
            "previous_training_time": 123.4,
            "train_iterator": {
                "version": 2,
                "epoch": 2,
                "iterations_in_epoch": 0,
                "shuffle": True,
                "len": 4,
            },
            "val_loss": 2.5,
            "best": 2.5,
        },
        # no "last_optimizer_state": fixture emulates --no-save-optimizer-state
    }
    os.makedirs(os.path.dirname(OUT), exist_ok=True)
    torch.save(state, OUT)
    print(f"wrote {OUT} ({os.path.getsize(OUT)} bytes)")


if __name__ == "__main__":
    main()
