#!/bin/bash
# Offline GEMM tuning for the Uni-Mol stress config (run on a GPU box).
# Its triangle/outer einsum GEMMs measured ~12 TF in the round-2 profile
# (profiles/unimol_bs1x8_r2.txt) and were never in the BERT-only tuned
# table. Same crash-isolated per-shape approach as tunableop_pershape.sh.
set -x
mkdir -p gpurun_out
EVO_ARGS="--task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
  --optimizer adam --adam-betas (0.9,0.99) --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 1e-4 --warmup-updates 1000 \
  --total-num-update 50000 --dataset-size 4800 \
  --batch-size 32 --atoms-per-mol 256 \
  --bf16 --log-interval 50 --log-format simple --no-save \
  --save-dir /tmp/ck_um"

# 1. record the untuned shapes over a few updates
rm -f gpurun_out/untuned_evo*.csv
PYTORCH_TUNABLEOP_ENABLED=1 \
PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_RECORD_UNTUNED=1 \
PYTORCH_TUNABLEOP_UNTUNED_FILENAME=gpurun_out/untuned_evo.csv \
timeout 300 python -m unicore_cli.train $EVO_ARGS --max-update 12 \
  > gpurun_out/evo_record.log 2>&1
SRC=$(ls gpurun_out/untuned_evo*.csv | head -1)
sort -u "$SRC" > gpurun_out/evo_shapes_all.csv
# batched (einsum/attention) shapes first — they are the measured 12 TF
# offenders; cap total so the sweep fits the box lease
{ grep -i "batched" gpurun_out/evo_shapes_all.csv || true; \
  grep -iv "batched" gpurun_out/evo_shapes_all.csv || true; } \
  > gpurun_out/evo_shapes.csv
wc -l gpurun_out/evo_shapes.csv

# 2. tune each shape in its own process (a segfault loses one shape)
rm -f gpurun_out/eshape_*.csv gpurun_out/evo_tune_status.log
i=0
while IFS= read -r line; do
  i=$((i+1))
  [ $i -gt 24 ] && break
  echo "$line" > "gpurun_out/eshape_${i}.csv"
  PYTORCH_TUNABLEOP_ENABLED=1 \
  PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME="gpurun_out/eshape_${i}_result.csv" \
  PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=50 \
  PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=30 \
  timeout 60 python -c "
import torch, torch.cuda.tunable as tunable
tunable.tune_gemm_in_file('gpurun_out/eshape_${i}.csv')
" >> gpurun_out/evo_pershape.log 2>&1 \
    && echo "shape $i OK: $line" >> gpurun_out/evo_tune_status.log \
    || echo "shape $i FAILED: $line" >> gpurun_out/evo_tune_status.log
done < gpurun_out/evo_shapes.csv
grep -c OK gpurun_out/evo_tune_status.log || true

# 3. merge results
OUT=gpurun_out/tuned_gemm_unimol.csv
rm -f "$OUT"
first=1
for f in gpurun_out/eshape_*_result*.csv; do
  [ -f "$f" ] || continue
  if [ $first -eq 1 ]; then cat "$f" >> "$OUT"; first=0
  else grep -v "^Validator" "$f" >> "$OUT" || true; fi
done
wc -l "$OUT" || true

# 4. same-box A/B
evo_ups () {
  timeout 400 python -m unicore_cli.train $EVO_ARGS --max-update 150 $1 2>&1 \
    | grep train_inner | tail -1 | grep -oE "ups=[0-9.]+"
}
echo "=== untuned:"; evo_ups ""
echo "=== tuned:";   evo_ups "--gemm-tuning-file $OUT"
echo "=== tuned:";   evo_ups "--gemm-tuning-file $OUT"
