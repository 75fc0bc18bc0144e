set -x
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/fap -o fa -- python tools/flash_microbench.py > gpurun_out/fa_prof.log 2>&1
echo RC=$?
f=$(find gpurun_out/fap -name "*kernel_stats.csv" | head -1)
python tools/top_kernels.py "$f" 12 > gpurun_out/fa_top.txt
rm -rf gpurun_out/fap
cat gpurun_out/fa_top.txt
