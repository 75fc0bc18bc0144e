set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
evorun () {
  timeout 360 python -m unicore_cli.train \
    --task evoformer_synthetic --arch evoformer --loss masked_msa \
    --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
    --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 1000 \
    --total-num-update 20000 --max-update 24 --dataset-size 256 \
    --batch-size 1 --update-freq 8 --msa-depth 128 --residues 256 \
    --bf16 --bf16-sr --ddp-backend c10d \
    --log-interval 8 --log-format simple --no-save \
    --save-dir /tmp/ck_evo 2>&1 | grep -E "train_inner" | tail -2
}
# GPU e2e tests for evoformer + the A/B
timeout 240 python -m pytest tests/ -q -m gpu -k "evo" 2>&1 | tail -1
echo "=== FOLD=0 (eager joins) ==="; UNICORE_FOLD_BIAS=0 evorun
echo "=== FOLD=1 (fused joins) ==="; evorun
