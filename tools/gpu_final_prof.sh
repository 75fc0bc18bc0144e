set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/pf -o fp -- \
  python bench.py --steps 5 --warmup 2 > gpurun_out/final_prof.log 2>&1
echo RC=$?
f=$(find gpurun_out/pf -name "*kernel_stats.csv" | head -1)
python - "$f" > gpurun_out/bert_final_top_kernels.txt <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
print(f"# total GPU kernel time over 7 fwd/bwd steps: {tot/1e9:.3f} s")
for r in rows[:30]:
    print(f'{100*float(r[key])/tot:7.3f}% {int(r["Calls"]):7d}x {float(r["AverageNs"])/1e3:10.2f}us  {r["Name"][:118]}')
PY
rm -rf gpurun_out/pf
head -6 gpurun_out/bert_final_top_kernels.txt
# eager self-baseline at the final config
timeout 350 python bench.py --steps 15 --warmup 4 --eager 2>/dev/null | grep -o 'value\": [0-9.]*'
timeout 350 python bench.py --steps 15 --warmup 4 2>/dev/null | grep -o 'value\": [0-9.]*'
