#!/usr/bin/env bash
# PMC counter evidence for the round-2 final build (run on an MI355X box).
# One counter group per pass (TCC slot budget: FETCH_SIZE=3, WRITE_SIZE=2
# can't share a pass); each pass its own rocprofv3 run over a short bench.
# Output: gpurun_out/bert_r2_pmc.txt (per-kernel HBM bytes + effective
# bandwidth + MFMA busy fraction for the top kernels).
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"

run_pmc () { # $1=counter $2=tag
  timeout 500 rocprofv3 --pmc "$1" --kernel-trace --stats \
    --output-format csv -d "gpurun_out/pmc_$2" -o "$2" -- \
    python bench.py --steps 3 --warmup 2 --no-eager-ab \
    > "gpurun_out/pmc_$2.log" 2>&1
  echo "pmc $1 rc=$?"
}

run_pmc FETCH_SIZE fetch
run_pmc WRITE_SIZE write
run_pmc SQ_VALU_MFMA_BUSY_CYCLES mfma

python - <<'PY' > gpurun_out/bert_r2_pmc.txt
import csv, glob, collections

def load(tag):
    # counter_collection.csv: one row per (dispatch, counter)
    vals = collections.defaultdict(float)
    for f in glob.glob(f"gpurun_out/pmc_{tag}/**/*counter_collection.csv",
                       recursive=True):
        for r in csv.DictReader(open(f)):
            name = r.get("Kernel_Name") or r.get("Kernel-Name") or ""
            v = r.get("Counter_Value") or r.get("Counter-Value") or 0
            vals[name.split("(")[0][:100]] += float(v)
    return vals

def durations(tag):
    d = collections.defaultdict(float)
    for f in glob.glob(f"gpurun_out/pmc_{tag}/**/*kernel_stats.csv",
                       recursive=True):
        for r in csv.DictReader(open(f)):
            d[r["Name"].split("(")[0][:100]] += float(r["TotalDurationNs"])
    return d

fetch, write, mfma = load("fetch"), load("write"), load("mfma")
dur = durations("fetch")
print("# PMC evidence, BERT-base bs127 seq512 bf16 bench (5 dispatig steps)")
print("# FETCH/WRITE = TCC fabric-side bytes (Infinity-Cache hits included);")
print("# eff GB/s = (FETCH+WRITE)/kernel-time from the same FETCH pass.")
print(f"{'kernel':<60} {'time_ms':>8} {'fetch_GB':>9} {'write_GB':>9} {'eff_TB/s':>9}")
for k, ns in sorted(dur.items(), key=lambda kv: -kv[1])[:18]:
    fb, wb = fetch.get(k, 0.0), write.get(k, 0.0)
    bw = (fb + wb) / ns if ns else 0.0  # bytes/ns = GB/s -> /1000 TB/s
    print(f"{k:<60.60} {ns/1e6:8.2f} {fb/2**30:9.2f} {wb/2**30:9.2f} {bw/1000:9.2f}")
mt = sum(mfma.values())
print(f"\n# total SQ_VALU_MFMA_BUSY_CYCLES over run: {mt:.3e}")
for k, v in sorted(mfma.items(), key=lambda kv: -kv[1])[:8]:
    print(f"  {v:.3e}  {k[:90]}")
PY
head -30 gpurun_out/bert_r2_pmc.txt
rm -rf gpurun_out/pmc_fetch gpurun_out/pmc_write gpurun_out/pmc_mfma
