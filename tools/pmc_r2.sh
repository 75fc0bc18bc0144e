#!/usr/bin/env bash
# PMC counter evidence for the round-2 final build (run on an MI355X box).
# One counter group per pass (TCC slot budget: FETCH_SIZE=3, WRITE_SIZE=2
# can't share a pass); each pass its own rocprofv3 run over a short bench.
# FETCH_SIZE/WRITE_SIZE report KB (classic rocprof convention; fabric-side
# TCC requests x 64/128 B, Infinity-Cache hits included).
# Output: gpurun_out/bert_r2_pmc.txt.
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
run_pmc MfmaUtil mfma

python - <<'PY' > gpurun_out/bert_r2_pmc.txt
import csv, glob, collections

def norm(name):
    name = name.replace("(anonymous namespace)::", "").replace("void ", "")
    return name.split("(")[0].split("<")[0][:64]

def load(tag, scale=1.0):
    vals = collections.defaultdict(float)
    n = collections.defaultdict(int)
    for f in glob.glob(f"gpurun_out/pmc_{tag}/**/*counter_collection.csv",
                       recursive=True):
        for r in csv.DictReader(open(f)):
            k = norm(r.get("Kernel_Name") or r.get("Kernel-Name") or "")
            v = float(r.get("Counter_Value") or r.get("Counter-Value") or 0)
            vals[k] += v * scale
            n[k] += 1
    return vals, n

def durations(tag):
    d = collections.defaultdict(float)
    for f in glob.glob(f"gpurun_out/pmc_{tag}/**/*kernel_stats.csv",
                       recursive=True):
        for r in csv.DictReader(open(f)):
            d[norm(r["Name"])] += float(r["TotalDurationNs"])
    return d

fetch, _ = load("fetch", 1024.0)   # KB -> bytes
write, _ = load("write", 1024.0)
mfma_sum, mfma_n = load("mfma")    # per-dispatch percent
dur = durations("fetch")
print("# PMC evidence, BERT-base bs127 seq512 bf16 bench (5 profiled steps)")
print("# fetch/write: TCC fabric-side bytes (FETCH_SIZE/WRITE_SIZE KB x 1024;")
print("# Infinity-Cache hits included). eff TB/s = (fetch+write)/time.")
print(f"{'kernel':<52} {'time_ms':>8} {'fetch_GB':>9} {'write_GB':>9} {'eff_TB/s':>9} {'MfmaUtil%':>10}")
for k, ns in sorted(dur.items(), key=lambda kv: -kv[1])[:20]:
    fb, wb = fetch.get(k, 0.0), write.get(k, 0.0)
    bw = (fb + wb) / ns / 1000.0 if ns else 0.0  # bytes/ns=GB/s; /1000=TB/s
    mu = mfma_sum.get(k, 0.0) / max(mfma_n.get(k, 1), 1)
    print(f"{k:<52.52} {ns/1e6:8.2f} {fb/2**30:9.2f} {wb/2**30:9.2f} {bw:9.2f} {mu:10.1f}")
PY
head -30 gpurun_out/bert_r2_pmc.txt
rm -rf gpurun_out/pmc_fetch gpurun_out/pmc_write gpurun_out/pmc_mfma
