"""End-to-end CPU training tests through the real CLI entry point
(BASELINE.json config 1: BERT MLM on CPU, world_size=1 — the plumbing test).

Covers: cold start, loss computation, checkpoint save, resume continuity
(iterator position + optimizer state + metrics restore), validation, EMA.
"""

import json
import os
import sys

import pytest
import torch

from unicore_cli import train as train_cli


BASE_ARGV = [
    "--task", "bert_synthetic",
    "--arch", "bert_base",
    "--loss", "masked_lm",
    "--optimizer", "adam",
    "--adam-betas", "(0.9, 0.98)",
    "--adam-eps", "1e-6",
    "--lr-scheduler", "polynomial_decay",
    "--total-num-update", "20",
    "--lr", "1e-3",
    "--batch-size", "4",
    "--dataset-size", "24",
    "--tokens-per-sample", "32",
    "--vocab-size", "128",
    "--encoder-layers", "2",
    "--encoder-embed-dim", "64",
    "--encoder-ffn-embed-dim", "128",
    "--encoder-attention-heads", "2",
    "--log-interval", "1",
    "--log-format", "simple",
    "--cpu",
    "--num-workers", "0",
    "--seed", "3",
]


def _run(argv, monkeypatch):
    monkeypatch.setattr(sys, "argv", ["unicore-train"] + argv)
    train_cli.cli_main()


def test_train_and_resume(tmp_path, monkeypatch):
    save_dir = str(tmp_path / "ckpt")
    argv = BASE_ARGV + [
        "--save-dir", save_dir,
        "--max-update", "4",
        "--save-interval-updates", "2",
        "--no-epoch-checkpoints",
    ]
    _run(argv, monkeypatch)
    assert os.path.exists(os.path.join(save_dir, "checkpoint_last.pt"))
    state = torch.load(
        os.path.join(save_dir, "checkpoint_last.pt"), weights_only=False
    )
    # checkpoint schema
    for key in ("args", "model", "optimizer_history", "extra_state",
                "last_optimizer_state", "task_state"):
        assert key in state, key
    assert state["extra_state"]["train_iterator"]["epoch"] >= 1
    hist = state["optimizer_history"][-1]
    assert hist["optimizer_name"] in ("Adam", "FusedAdam", "UnicoreAdam")
    assert hist["num_updates"] == 4

    # resume: must continue from update 4, not restart
    argv2 = BASE_ARGV + [
        "--save-dir", save_dir,
        "--max-update", "8",
        "--save-interval-updates", "2",
        "--no-epoch-checkpoints",
    ]
    _run(argv2, monkeypatch)
    state2 = torch.load(
        os.path.join(save_dir, "checkpoint_last.pt"), weights_only=False
    )
    assert state2["optimizer_history"][-1]["num_updates"] == 8
    # metrics meters restored and advanced
    assert "metrics" in state2["extra_state"]


def test_validate_and_best_checkpoint(tmp_path, monkeypatch):
    save_dir = str(tmp_path / "ckpt")
    argv = BASE_ARGV + [
        "--save-dir", save_dir,
        "--max-epoch", "2",
        "--best-checkpoint-metric", "loss",
    ]
    _run(argv, monkeypatch)
    assert os.path.exists(os.path.join(save_dir, "checkpoint_best.pt"))
    assert os.path.exists(os.path.join(save_dir, "checkpoint_last.pt"))
    assert os.path.exists(os.path.join(save_dir, "checkpoint2.pt"))


def test_grad_accumulation_equivalence(tmp_path, monkeypatch):
    """update-freq 2 with batch 2 must track batch 4 within fp32 tolerance."""
    import numpy as np
    from unicore_amd import options, tasks, utils
    from unicore_amd.trainer import Trainer

    def build(bsz, update_freq):
        parser = options.get_training_parser()
        args = options.parse_args_and_arch(
            parser,
            input_args=BASE_ARGV
            + ["--batch-size", str(bsz), "--update-freq", str(update_freq),
               "--dropout", "0", "--attention-dropout", "0",
               "--activation-dropout", "0", "--emb-dropout", "0",
               "--save-dir", str(tmp_path / f"uf{update_freq}")],
        )
        args.distributed_world_size = 1
        args.distributed_rank = 0
        args.device_id = 0
        torch.manual_seed(args.seed)
        np.random.seed(args.seed)
        task = tasks.setup_task(args)
        model = task.build_model(args)
        loss = task.build_loss(args)
        trainer = Trainer(args, task, model, loss)
        epoch_itr = trainer.get_train_iterator(epoch=1)
        trainer.init_total_train_steps(epoch_itr)
        return args, trainer, epoch_itr

    def steps(trainer, epoch_itr, update_freq, n_updates):
        itr = epoch_itr.next_epoch_itr(shuffle=False)
        from unicore_amd.data import iterators

        grouped = iterators.GroupedIterator(itr, update_freq)
        logs = []
        for i, samples in enumerate(grouped):
            if i >= n_updates:
                break
            out = trainer.train_step(samples)
            logs.append(out)
        return trainer

    _, tr_big, it_big = build(8, 1)
    tr_big = steps(tr_big, it_big, 1, 2)
    _, tr_acc, it_acc = build(4, 2)
    tr_acc = steps(tr_acc, it_acc, 2, 2)

    p_big = list(tr_big.model.parameters())
    p_acc = list(tr_acc.model.parameters())
    diffs = [(a - b).abs().max().item() for a, b in zip(p_big, p_acc)]
    assert max(diffs) < 5e-4, max(diffs)


def test_ema_training(tmp_path, monkeypatch):
    save_dir = str(tmp_path / "ckpt")
    argv = BASE_ARGV + [
        "--save-dir", save_dir,
        "--max-update", "3",
        "--validate-with-ema",
        "--ema-decay", "0.9",
    ]
    _run(argv, monkeypatch)
    state = torch.load(
        os.path.join(save_dir, "checkpoint_last.pt"), weights_only=False
    )
    assert "ema" in state and state["ema"] is not None


def test_evoformer_recycling_and_activation_checkpoint(tmp_path):
    """Uni-Fold-style recycling (no-grad passes + one grad pass) and
    per-block activation checkpointing both train on CPU."""
    from unicore_cli.train import cli_main
    import sys

    argv = [
        "train.py", "--task", "evoformer_synthetic", "--arch", "evoformer",
        "--loss", "masked_msa", "--optimizer", "adam",
        "--lr-scheduler", "fixed", "--lr", "1e-4",
        "--max-update", "2", "--dataset-size", "8", "--batch-size", "2",
        "--msa-depth", "8", "--residues", "16", "--evo-layers", "2",
        "--msa-dim", "32", "--pair-dim", "16", "--evo-heads", "4",
        "--recycle-iters", "1", "--activation-checkpoint",
        "--cpu", "--log-interval", "1", "--log-format", "simple",
        "--no-save", "--save-dir", str(tmp_path), "--num-workers", "0",
    ]
    old = sys.argv
    sys.argv = argv
    try:
        cli_main()
    finally:
        sys.argv = old


def test_infer_demo_roundtrip(tmp_path):
    """Train 2 updates, save, then run the inference demo on the saved
    checkpoint (CPU)."""
    import subprocess
    import sys
    import os

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "unicore_cli.train",
         "--task", "bert_synthetic", "--arch", "bert_base",
         "--encoder-layers", "2", "--encoder-embed-dim", "64",
         "--encoder-ffn-embed-dim", "128", "--encoder-attention-heads", "4",
         "--loss", "masked_lm", "--optimizer", "adam",
         "--lr-scheduler", "fixed", "--lr", "1e-4",
         "--max-update", "2", "--dataset-size", "8", "--batch-size", "2",
         "--tokens-per-sample", "64", "--max-seq-len", "66",
         "--vocab-size", "100", "--cpu", "--log-format", "simple",
         "--num-workers", "0", "--save-dir", str(tmp_path)],
        cwd=repo, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    ck = tmp_path / "checkpoint_last.pt"
    assert ck.exists()
    r = subprocess.run(
        [sys.executable, "examples/bert/infer_demo.py", "--checkpoint",
         str(ck), "--cpu", "--batch-size", "2", "--seq-len", "64",
         "--iters", "2"],
        cwd=repo, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    assert "tokens/s" in r.stdout
