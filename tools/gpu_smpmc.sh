set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_VALU SQ_ACTIVE_INST_VMEM SQ_INSTS_VALU \
  --output-format csv -d gpurun_out/pmc -o sp -- python bench.py --steps 3 --warmup 1 > gpurun_out/sm_pmc.log 2>&1
echo RC=$?
f=$(find gpurun_out/pmc -name "*counter_collection.csv" | head -1)
python - "$f" <<'PY'
import csv, sys, collections, re
rows = list(csv.DictReader(open(sys.argv[1])))
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for r in rows:
    n = r.get("Kernel_Name") or ""
    m = re.search(r"(softmax_fwd_vec|softmax_bwd_biasgrad|gelu_dropout_bwd|qkv_split_bwd)", n)
    if not m: continue
    agg[m.group(1)][r["Counter_Name"]] += float(r["Counter_Value"])
for n, c in agg.items():
    wc = c.get("SQ_WAVE_CYCLES", 1)
    print(n)
    for k, v in sorted(c.items()):
        print(f"  {k:24s} {v/1e9:8.2f}G ({100*v/wc:5.1f}%)")
PY
rm -rf gpurun_out/pmc
