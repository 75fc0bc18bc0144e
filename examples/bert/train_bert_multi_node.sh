#!/usr/bin/env bash
# Multi-node BERT pretraining over RCCL (xGMI intra-node, network inter-node).
# Analog of the reference's examples/bert/train_bert_test_multi_node.sh:
# launch this script on every node with the env below set (or under Slurm,
# where unicore-train infers ranks itself via --distributed-port).
#
#   MASTER_ADDR=<node0 hostname/ip>  MASTER_PORT=29500
#   NNODES=<total nodes>             NODE_RANK=<this node's index>
#   GPUS_PER_NODE=8
#
# NCCL_ASYNC_ERROR_HANDLING makes RCCL surface communicator failures as
# exceptions instead of hangs (the reference sets the same).
set -e
export NCCL_ASYNC_ERROR_HANDLING=1
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

GPUS_PER_NODE=${GPUS_PER_NODE:-8}
NNODES=${NNODES:-1}
NODE_RANK=${NODE_RANK:-0}
MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
MASTER_PORT=${MASTER_PORT:-29500}

exec python -m torch.distributed.run \
  --nnodes "$NNODES" --node-rank "$NODE_RANK" \
  --nproc-per-node "$GPUS_PER_NODE" \
  --master-addr "$MASTER_ADDR" --master-port "$MASTER_PORT" \
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
  --save-dir ./checkpoints_bert "$@"
