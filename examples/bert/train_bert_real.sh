#!/bin/bash
# Real-corpus BERT pretraining, the reference workflow end to end:
#   1. corpus -> data dir (WordPiece vocab + stored splits)
#   2. torchrun data-parallel training via the unicore-train CLI
# Usage: bash examples/bert/train_bert_real.sh <corpus.txt> <data_dir> [n_gpus]
set -e
CORPUS=${1:-examples/bert/sample_corpus.txt}
DATA=${2:-./bert_corpus_data}
NGPU=${3:-1}

[ -f "$DATA/dict.txt" ] || python examples/bert/prepare_corpus.py "$CORPUS" \
    --out-dir "$DATA" --vocab-size 8000

python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPU" \
    --master-addr 127.0.0.1 --master-port 29600 \
    -m unicore_cli.train \
    "$DATA" \
    --task bert --arch bert_base --loss masked_lm \
    --optimizer adam --adam-betas '(0.9, 0.98)' --adam-eps 1e-6 \
    --lr-scheduler polynomial_decay --lr 1e-4 --total-num-update 100000 \
    --warmup-updates 1000 --clip-norm 1.0 \
    --batch-size 32 --max-seq-len 512 \
    --bf16 --max-update 100000 --save-interval-updates 1000 \
    --save-dir ./bert_real_ckpt --log-format simple --log-interval 50 \
    "${@:4}"
