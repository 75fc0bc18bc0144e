set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_kernels_gpu.py -k gaussian -q > gpurun_out/gauss_tests.log 2>&1
echo "TESTS_RC=$?"; tail -2 gpurun_out/gauss_tests.log
molrun () {
  timeout 360 python -m unicore_cli.train \
    --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
    --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
    --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
    --total-num-update 50000 --max-update 40 --dataset-size 2560 \
    --batch-size 32 --atoms-per-mol 256 --bf16 --ddp-backend c10d \
    --log-interval 10 --log-format simple --no-save \
    --save-dir /tmp/ck_mol 2>&1 | grep -E "train_inner" | tail -2
}
echo "=== UNFUSED-BASIS (prev best) ==="; UNICORE_GAUSSIAN_EAGER=1 molrun
echo "=== FULLY-FUSED ==="; molrun
# profile fully-fused
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof_mol -o molfused -- \
  python -m unicore_cli.train \
    --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
    --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
    --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
    --total-num-update 50000 --max-update 15 --dataset-size 960 \
    --batch-size 32 --atoms-per-mol 256 --bf16 --ddp-backend c10d \
    --log-interval 5 --log-format simple --no-save \
    --save-dir /tmp/ck_mol > gpurun_out/mol_prof_run.log 2>&1
f=$(find gpurun_out/prof_mol -name "*kernel_stats.csv" | head -1)
if [ -n "$f" ]; then
  python - "$f" > gpurun_out/mol_fused_top_kernels.txt <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
for r in rows[:30]:
    print(f'{100*float(r[key])/tot:7.3f}% {int(r["Calls"]):7d}x {float(r["AverageNs"])/1e3:10.2f}us  {r["Name"][:120]}')
PY
fi
rm -rf gpurun_out/prof_mol
head -10 gpurun_out/mol_fused_top_kernels.txt
