#!/usr/bin/env python3
"""Corpus -> BERT pretraining data directory, matching the reference's LMDB
workflow (reference examples/bert/task.py:31-124: {split}.lmdb + dict.txt).

Steps:
  1. read raw text file(s), one training example per non-empty line;
  2. train a WordPiece vocabulary on the corpus (HuggingFace tokenizers)
     and write it as dict.txt;
  3. split lines into train/valid and store each split as {split}.lmdb
     (when the lmdb package is installed) or {split}.kv (our dependency-free
     single-file store; the bert task reads either).

Usage:
  python examples/bert/prepare_corpus.py corpus.txt [corpus2.txt ...] \
      --out-dir data/bert_corpus --vocab-size 8000 --valid-fraction 0.05
"""

import argparse
import os
import random
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), "..", ".."))
sys.path.insert(0, REPO)


def read_lines(paths, min_chars):
    lines = []
    for path in paths:
        with open(path, "r", encoding="utf-8") as f:
            for raw in f:
                text = raw.strip()
                if len(text) >= min_chars:
                    lines.append(text)
    return lines


def train_wordpiece(paths, out_dir, vocab_size):
    from tokenizers import BertWordPieceTokenizer

    tok = BertWordPieceTokenizer(lowercase=True)
    tok.train(
        files=list(paths),
        vocab_size=vocab_size,
        special_tokens=["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"],
    )
    vocab_file = os.path.join(out_dir, "dict.txt")
    # dict.txt = one token per line, index order (tokenizers vocab format)
    ordered = sorted(tok.get_vocab().items(), key=lambda kv: kv[1])
    with open(vocab_file, "w", encoding="utf-8") as f:
        for token, _ in ordered:
            f.write(token + "\n")
    return vocab_file


def write_split(out_dir, split, lines):
    try:
        import lmdb

        path = os.path.join(out_dir, f"{split}.lmdb")
        env = lmdb.open(path, subdir=False, map_size=1 << 32)
        with env.begin(write=True) as txn:
            for i, text in enumerate(lines):
                import pickle

                txn.put(str(i).encode(), pickle.dumps(text))
        env.close()
    except ImportError:
        from unicore_amd.data.kv_dataset import KVWriter

        path = os.path.join(out_dir, f"{split}.kv")
        with KVWriter(path) as w:
            for text in lines:
                w.put(text)
    print(f"{split}: {len(lines)} examples -> {path}")
    return path


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("inputs", nargs="+", help="raw text files (one example per line)")
    ap.add_argument("--out-dir", required=True)
    ap.add_argument("--vocab-size", type=int, default=8000)
    ap.add_argument("--valid-fraction", type=float, default=0.05)
    ap.add_argument("--min-chars", type=int, default=16)
    ap.add_argument("--seed", type=int, default=1)
    args = ap.parse_args()

    os.makedirs(args.out_dir, exist_ok=True)
    lines = read_lines(args.inputs, args.min_chars)
    assert lines, "no usable lines in the corpus"
    print(f"corpus: {len(lines)} lines from {len(args.inputs)} file(s)")

    train_wordpiece(args.inputs, args.out_dir, args.vocab_size)

    rng = random.Random(args.seed)
    rng.shuffle(lines)
    n_valid = max(1, int(len(lines) * args.valid_fraction))
    write_split(args.out_dir, "valid", lines[:n_valid])
    write_split(args.out_dir, "train", lines[n_valid:])
    print(f"data dir ready: {args.out_dir}")


if __name__ == "__main__":
    main()
