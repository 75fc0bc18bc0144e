set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
# 1. parity + determinism tests
timeout 300 python -m pytest tests/test_kernels_gpu.py -k gaussian -x -q > gpurun_out/gauss_tests.log 2>&1
echo "TESTS_RC=$?"
tail -3 gpurun_out/gauss_tests.log

molrun () {
  timeout 360 python -m unicore_cli.train \
    --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
    --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
    --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
    --total-num-update 50000 --max-update 40 --dataset-size 2560 \
    --batch-size 32 --atoms-per-mol 256 --bf16 --ddp-backend c10d \
    --log-interval 10 --log-format simple --no-save \
    --save-dir /tmp/ck_mol 2>&1 | grep -E "train_inner|done training" | tail -5
}
# 2. A/B step timing (eager chain vs fused kernel)
echo "=== EAGER ==="; UNICORE_GAUSSIAN_EAGER=1 molrun
echo "=== FUSED ==="; molrun
