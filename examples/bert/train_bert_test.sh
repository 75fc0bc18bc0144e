#!/usr/bin/env bash
# Single-node BERT-base MLM pretraining on N MI355X GPUs over RCCL/xGMI.
# Synthetic-data analog of the reference's examples/bert/train_bert_test.sh
# (swap --task bert_synthetic for --task bert + an LMDB data dir to train on
# real tokenized text).
set -e
n_gpu=${1:-8}
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n_gpu" \
  --master-addr 127.0.0.1 --master-port 29500 \
  -m unicore_cli.train \
  --task bert_synthetic --arch bert_base --loss masked_lm \
  --optimizer adam --adam-betas '(0.9, 0.98)' --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
  --total-num-update 100000 --max-update 100000 \
  --batch-size 96 --tokens-per-sample 512 --max-seq-len 514 \
  --dataset-size 100000 \
  --bf16 --ddp-backend c10d --bucket-cap-mb 32 \
  --log-interval 50 --log-format simple \
  --save-interval-updates 1000 --keep-interval-updates 5 --no-epoch-checkpoints \
  --save-dir ./checkpoints_bert "${@:2}"
