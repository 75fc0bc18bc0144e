#!/usr/bin/env bash
# Evoformer-block masked-MSA pretraining, bf16 + stochastic rounding +
# grad-accum 8 — BASELINE.json stress config 5.
set -e
n_gpu=${1:-8}
# offline-tuned hipBLASLt/rocBLAS algorithms for this config's batched
# einsum GEMMs (+2.3% measured; see tools/tune_evoformer.sh)
TUNED="$(dirname "$0")/../../tools/tuned_gemm_evoformer.csv"
EXTRA=()
[ -f "$TUNED" ] && EXTRA=(--gemm-tuning-file "$TUNED")
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$n_gpu" \
  --master-addr 127.0.0.1 --master-port 29502 \
  -m unicore_cli.train \
  --task evoformer_synthetic --arch evoformer --loss masked_msa \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
  --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 1000 \
  --total-num-update 20000 --max-update 20000 \
  --batch-size 1 --update-freq 8 --msa-depth 128 --residues 256 \
  --bf16 --bf16-sr --ddp-backend c10d \
  --log-interval 10 --log-format simple \
  --save-dir ./checkpoints_evo "${EXTRA[@]}" "${@:2}"
