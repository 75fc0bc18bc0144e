set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
# 1. full GPU test suite (all files)
timeout 900 python -m pytest tests/ -q -m gpu > gpurun_out/full_gpu_tests.log 2>&1
echo "SUITE_RC=$?"
grep -E "^[0-9]+ (passed|failed)|passed|failed" gpurun_out/full_gpu_tests.log | tail -1
# 2. smoke
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" > gpurun_out/smoke.log 2>&1
echo "SMOKE_RC=$?"; tail -1 gpurun_out/smoke.log
# 3. default bench (what the driver runs)
timeout 600 python bench.py --steps 20 --warmup 5 > gpurun_out/bench_default.log 2>&1
echo "BENCH_RC=$?"; grep '"metric"' gpurun_out/bench_default.log | tail -1
