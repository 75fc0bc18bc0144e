#!/bin/bash
# Record + per-shape-tune the GEMMs of a given bench batch size, then A/B.
# Usage (on a GPU box): bash tools/tune_batch.sh 128
set -x
BS=${1:-128}
mkdir -p gpurun_out

# stage 1: record untuned shapes at this batch size
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_RECORD_UNTUNED=1 \
PYTORCH_TUNABLEOP_UNTUNED_FILENAME=gpurun_out/untuned_bs${BS}.csv \
UNICORE_NO_TUNED_GEMM=1 \
timeout 300 python bench.py --steps 3 --warmup 2 --batch-size ${BS} \
  --no-eager-ab > gpurun_out/tune_bs${BS}_s1.log 2>&1

# stage 2: per-shape isolated tuning of the stable hot shapes
ROWS=$((BS * 512))
grep -hE "${ROWS}|B_$((BS * 12))" gpurun_out/untuned_bs${BS}*.csv \
  | sed 's/^[^:]*://' | sort -u > gpurun_out/stable_bs${BS}.csv
wc -l gpurun_out/stable_bs${BS}.csv
i=0
while IFS= read -r line; do
  i=$((i+1))
  echo "$line" > "gpurun_out/bs${BS}_shape_${i}.csv"
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME="gpurun_out/bs${BS}_shape_${i}_result.csv" \
  PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=50 \
  PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=30 \
  timeout 120 python -c "
import torch, torch.cuda.tunable as tunable
tunable.tune_gemm_in_file('gpurun_out/bs${BS}_shape_${i}.csv')
" >> gpurun_out/bs${BS}_tune.log 2>&1 \
    && echo "shape $i OK: $line" >> gpurun_out/bs${BS}_status.log \
    || echo "shape $i FAILED: $line" >> gpurun_out/bs${BS}_status.log
done < gpurun_out/stable_bs${BS}.csv
cat gpurun_out/bs${BS}_status.log

# stage 3: merge (existing tuned table + new shapes) and A/B at this batch
python - <<PYEOF
from pathlib import Path
out, seen_validator = [], False
srcs = [Path('tools/tuned_gemm_bert.csv')] if Path('tools/tuned_gemm_bert.csv').exists() else []
srcs += sorted(Path('gpurun_out').glob('bs${BS}_shape_*_result*.csv'))
for f in srcs:
    for line in f.read_text().splitlines():
        if line.startswith('Validator'):
            if not seen_validator:
                out.append(line)
        elif line.strip():
            out.append(line)
    seen_validator = True
Path('gpurun_out/tuned_gemm_bs${BS}.csv').write_text('\n'.join(out) + '\n')
print('merged', len(out), 'lines')
PYEOF

UNICORE_NO_TUNED_GEMM=1 PYTORCH_TUNABLEOP_ENABLED=1 \
PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tuned_gemm_bs${BS}.csv \
timeout 420 python bench.py --steps 15 --warmup 5 --batch-size ${BS} \
  --no-eager-ab > gpurun_out/bench_bs${BS}_tuned.log 2>&1
UNICORE_NO_TUNED_GEMM=1 \
timeout 420 python bench.py --steps 15 --warmup 5 --batch-size ${BS} \
  --no-eager-ab > gpurun_out/bench_bs${BS}_plain.log 2>&1
echo "BS=${BS} TUNED:"; tail -1 gpurun_out/bench_bs${BS}_tuned.log
echo "BS=${BS} PLAIN:"; tail -1 gpurun_out/bench_bs${BS}_plain.log
