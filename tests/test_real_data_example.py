"""End-to-end real-corpus workflow, matching the reference's example flow
(reference examples/bert/train_bert_test.sh + examples/bert/task.py:31-124):
raw text -> prepare_corpus.py (WordPiece vocab + stored splits) -> the
built-in ``bert`` task -> unicore-train CLI -> checkpoint on disk.

Runs on CPU with a tiny model; uses the .kv storage backend when lmdb is
not installed (the prep script picks automatically).
"""

import json
import os
import subprocess
import sys

import torch

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
CORPUS = os.path.join(REPO, "examples", "bert", "sample_corpus.txt")


def _prepare(tmp_path):
    data_dir = str(tmp_path / "corpus_data")
    subprocess.check_call(
        [
            sys.executable,
            os.path.join(REPO, "examples", "bert", "prepare_corpus.py"),
            CORPUS,
            "--out-dir", data_dir,
            "--vocab-size", "1500",
            "--valid-fraction", "0.1",
        ],
        cwd=REPO,
    )
    return data_dir


def test_prepare_and_train_real_corpus(tmp_path):
    data_dir = _prepare(tmp_path)
    for artifact in ("dict.txt",):
        assert os.path.isfile(os.path.join(data_dir, artifact))
    assert any(
        os.path.isfile(os.path.join(data_dir, f"train{ext}"))
        for ext in (".lmdb", ".kv")
    )

    save_dir = str(tmp_path / "ckpt")
    from unicore_cli.train import main as train_main
    from unicore_amd import options

    argv = [
        data_dir,
        "--task", "bert",
        "--arch", "bert_base",
        "--loss", "masked_lm",
        "--optimizer", "adam",
        "--lr-scheduler", "fixed",
        "--lr", "1e-3",
        "--batch-size", "4",
        "--max-seq-len", "128",
        "--encoder-layers", "2",
        "--encoder-embed-dim", "64",
        "--encoder-ffn-embed-dim", "128",
        "--encoder-attention-heads", "2",
        "--max-update", "6",
        "--log-format", "json",
        "--log-interval", "2",
        "--num-workers", "0",
        "--save-dir", save_dir,
        "--tmp-save-dir", save_dir,
        "--seed", "4",
        "--cpu",
        "--disable-validation",
    ]
    parser = options.get_training_parser()
    args = options.parse_args_and_arch(parser, input_args=argv)
    args.distributed_world_size = 1
    args.distributed_rank = 0
    args.device_id = 0
    args.distributed_no_spawn = True
    train_main(args)

    # a checkpoint landed and it reloads with the real vocabulary
    last = os.path.join(save_dir, "checkpoint_last.pt")
    assert os.path.isfile(last)
    state = torch.load(last, map_location="cpu", weights_only=False)
    assert state["optimizer_history"][-1]["num_updates"] >= 6
    emb = state["model"]["embed_tokens.weight"]
    assert emb.shape[0] >= 1000  # trained WordPiece vocab
