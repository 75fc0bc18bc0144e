set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/pf -o fb -- python tools/flash_microbench.py > gpurun_out/flash_prof.log 2>&1
echo RC=$?
f=$(find gpurun_out/pf -name "*kernel_stats.csv" | head -1)
python - "$f" <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
for r in sorted(rows, key=lambda r: -float(r["TotalDurationNs"])):
    n = r["Name"]
    if any(s in n for s in ("flash", "softmax", "Cijk", "dropout")):
        print(f'{int(r["Calls"]):5d}x {float(r["AverageNs"])/1e3:9.2f}us  {n[:100]}')
PY
rm -rf gpurun_out/pf
tail -6 gpurun_out/flash_prof.log
