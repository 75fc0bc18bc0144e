set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/pb -o bp -- \
  python bench.py --steps 5 --warmup 2 > gpurun_out/bert_prof.log 2>&1
echo RC=$?
f=$(find gpurun_out/pb -name "*kernel_stats.csv" | head -1)
python - "$f" > gpurun_out/bert_top_kernels_biasfold.txt <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
for r in rows[:34]:
    print(f'{100*float(r[key])/tot:7.3f}% {int(r["Calls"]):7d}x {float(r["AverageNs"])/1e3:10.2f}us  {r["Name"][:118]}')
PY
rm -rf gpurun_out/pb
head -34 gpurun_out/bert_top_kernels_biasfold.txt
