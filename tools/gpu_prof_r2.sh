set -x
cd $GRAFT_REPO_ROOT
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/pf2 -o r2 -- \
  python bench.py --steps 5 --warmup 2 --no-eager-ab > gpurun_out/r2_prof.log 2>&1
echo RC=$?
f=$(find gpurun_out/pf2 -name "*kernel_stats.csv" | head -1)
python - "$f" > gpurun_out/bert_r2_top_kernels.txt <<'PY'
import csv, sys
rows = list(csv.DictReader(open(sys.argv[1])))
key = "TotalDurationNs"
rows.sort(key=lambda r: -float(r[key]))
tot = sum(float(r[key]) for r in rows)
print("# BERT-base bs127 seq512 bf16, round 2 (tuned GEMMs + fused LN join)")
print(f"# total GPU kernel time over 7 fwd/bwd steps: {tot/1e9:.3f} s")
for r in rows[:30]:
    print(f'{100*float(r[key])/tot:7.3f}% {int(r["Calls"]):7d}x {float(r["AverageNs"])/1e3:10.2f}us  {r["Name"][:118]}')
PY
rm -rf gpurun_out/pf2
head -8 gpurun_out/bert_r2_top_kernels.txt
# evoformer hipGraph A/B + parity test
timeout 300 python -m pytest tests/test_stress_models.py -k hip_graph -x -q > gpurun_out/evo_graph_test.log 2>&1; tail -2 gpurun_out/evo_graph_test.log
evorun () {
  timeout 360 python -m unicore_cli.train \
    --task evoformer_synthetic --arch evoformer --loss masked_msa \
    --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
    --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 1000 \
    --total-num-update 20000 --max-update 64 --dataset-size 256 \
    --batch-size 1 --update-freq 8 --msa-depth 128 --residues 256 \
    --dropout 0.0 $1 \
    --bf16 --bf16-sr --ddp-backend c10d \
    --log-interval 32 --log-format simple --no-save \
    --save-dir /tmp/ck_evo 2>&1 | grep -E "train_inner" | tail -1
}
echo "=== EVO plain (dropout 0) ==="; evorun ""
echo "=== EVO hipGraph blocks ==="; evorun "--hip-graph-blocks"
