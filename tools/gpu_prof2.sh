set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
prof () {
  timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/p1 -o z -- "$@" > gpurun_out/prof_run.log 2>&1
  f=$(find gpurun_out/p1 -name "*kernel_stats.csv" | head -1)
  python - "$f" <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
for r in rows[:26]:
    print(f'{100*float(r[key])/tot:7.3f}% {int(r["Calls"]):7d}x {float(r["AverageNs"])/1e3:10.2f}us  {r["Name"][:116]}')
PY
  rm -rf gpurun_out/p1
}
prof python -m unicore_cli.train \
  --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
  --total-num-update 50000 --max-update 15 --dataset-size 960 \
  --batch-size 32 --atoms-per-mol 256 --bf16 --ddp-backend c10d \
  --log-interval 5 --log-format simple --no-save --save-dir /tmp/c1 > gpurun_out/mol_final_top_kernels.txt
head -3 gpurun_out/mol_final_top_kernels.txt
prof python -m unicore_cli.train \
  --task evoformer_synthetic --arch evoformer --loss masked_msa \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
  --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 1000 \
  --total-num-update 20000 --max-update 16 --dataset-size 192 \
  --batch-size 1 --update-freq 8 --msa-depth 128 --residues 256 \
  --bf16 --bf16-sr --ddp-backend c10d --log-interval 8 --log-format simple \
  --no-save --save-dir /tmp/c2 > gpurun_out/evo_final_top_kernels.txt
head -3 gpurun_out/evo_final_top_kernels.txt
