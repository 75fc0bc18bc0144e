#!/usr/bin/env bash
# Uni-Mol-style 3D molecular pretraining (pair-bias attention + RMSNorm +
# coordinate denoising) on N MI355X GPUs — BASELINE.json stress config 4.
set -e
n_gpu=${1:-8}
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n_gpu" \
  --master-addr 127.0.0.1 --master-port 29501 \
  -m unicore_cli.train \
  --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
  --total-num-update 50000 --max-update 50000 \
  --batch-size 32 --atoms-per-mol 256 \
  --bf16 --ddp-backend c10d \
  --log-interval 50 --log-format simple \
  --save-dir ./checkpoints_unimol "${@:2}"
