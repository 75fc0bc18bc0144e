set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_WAIT_INST_LDS SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_ACTIVE_INST_LDS SQ_ACTIVE_INST_VMEM \
  --output-format csv -d gpurun_out/pmc -o fb -- python tools/flash_microbench.py > gpurun_out/flash_pmc.log 2>&1
echo RC=$?
f=$(find gpurun_out/pmc -name "*counter_collection.csv" | head -1)
python - "$f" <<'PY'
import csv, sys, collections, re
rows = list(csv.DictReader(open(sys.argv[1])))
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for r in rows:
    n = r.get("Kernel_Name") or ""
    m = re.search(r"(flash_\w+_kernel|flash_dot\w*)", n)
    if not m: continue
    agg[m.group(1)][r["Counter_Name"]] += float(r["Counter_Value"])
for n, c in agg.items():
    wc = c.get("SQ_WAVE_CYCLES", 1)
    print(n)
    for k, v in sorted(c.items()):
        print(f"  {k:28s} {v/1e9:8.3f}G  ({100*v/wc:5.1f}%)")
PY
rm -rf gpurun_out/pmc
