#!/bin/bash
# Offline hipBLASLt GEMM tuning pipeline for the BERT bench (run on a GPU box).
# Stage 1: record the untuned GEMM shapes from a short bench run.
# Stage 2: tune them offline (separate process; a crash only loses the tuning).
# Stage 3: A/B the bench with the tuned results loaded.
set -x
mkdir -p gpurun_out

# Stage 1: record untuned shapes (no tuning -> no crash risk)
PYTORCH_TUNABLEOP_ENABLED=1 \
PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_RECORD_UNTUNED=1 \
PYTORCH_TUNABLEOP_UNTUNED_FILENAME=gpurun_out/untuned_gemm.csv \
timeout 300 python bench.py --steps 3 --warmup 2 --no-eager-ab \
  > gpurun_out/tune_stage1.log 2>&1
wc -l gpurun_out/untuned_gemm.csv* || true

# Stage 2: offline tuning in its own process
PYTORCH_TUNABLEOP_ENABLED=1 \
PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tuned_gemm.csv \
PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=120 \
PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=60 \
timeout 900 python - <<'PYEOF' > gpurun_out/tune_stage2.log 2>&1
import glob
import torch
import torch.cuda.tunable as tunable
files = sorted(glob.glob("gpurun_out/untuned_gemm*.csv"))
print("untuned files:", files)
for f in files:
    tunable.tune_gemm_in_file(f)
print("tuning done; results:", tunable.get_filename())
PYEOF
tail -3 gpurun_out/tune_stage2.log
ls -la gpurun_out/tuned_gemm* || true

# Stage 3: A/B with tuned results (if produced)
if ls gpurun_out/tuned_gemm*.csv >/dev/null 2>&1; then
  PYTORCH_TUNABLEOP_ENABLED=1 \
  PYTORCH_TUNABLEOP_TUNING=0 \
  PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tuned_gemm.csv \
  timeout 420 python bench.py --steps 15 --warmup 5 --no-eager-ab \
    > gpurun_out/tune_stage3_tuned.log 2>&1
  timeout 420 python bench.py --steps 15 --warmup 5 --no-eager-ab \
    > gpurun_out/tune_stage3_base.log 2>&1
  echo TUNED:; tail -1 gpurun_out/tune_stage3_tuned.log
  echo BASE:;  tail -1 gpurun_out/tune_stage3_base.log
fi
