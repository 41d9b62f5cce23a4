#!/bin/bash
# Round-2 GPU call 1: carry-over verification (ROADMAP §3c)
#  1. full GPU test suite (validates late r01 kernel changes)
#  2. headline 160L bench (graphed) + fp32 24L datapoint
#  3. rocprofv3 kernel-stats profile of the bench
#  4. hand-GEMM baseline table (input to the 8-phase GEMM work)
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/r02_pytest_gpu.log 2>&1
echo "pytest rc=$?" >> gpurun_out/r02_pytest_gpu.log
tail -4 gpurun_out/r02_pytest_gpu.log

timeout 600 python bench.py --steps 20 --warmup 3 \
  --json-out gpurun_out/r02_b160.json > gpurun_out/r02_b160.log 2>&1
echo "bench160 rc=$?"; tail -2 gpurun_out/r02_b160.log

timeout 600 python bench.py --layers 24 --dtype fp32 --steps 10 --warmup 3 \
  --json-out gpurun_out/r02_b24_fp32.json > gpurun_out/r02_b24_fp32.log 2>&1
echo "bench24fp32 rc=$?"; tail -2 gpurun_out/r02_b24_fp32.log

cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r02_prof \
  -o r02b160 -- python /root/repo/bench.py --steps 3 --warmup 1 \
  > /root/repo/gpurun_out/r02_prof.log 2>&1
echo "rocprof rc=$?"
cd /root/repo
DB=$(ls gpurun_out/r02_prof/*.db 2>/dev/null | head -1)
if [ -n "$DB" ]; then
  python tools/prof_summary.py "$DB" 60 gpurun_out/r02_b160_kernels.txt \
    >> gpurun_out/r02_prof.log 2>&1
  head -30 gpurun_out/r02_b160_kernels.txt
fi

timeout 400 python tools/gemm_bench.py > gpurun_out/r02_gemm_baseline.txt 2>&1
echo "gemm rc=$?"; cat gpurun_out/r02_gemm_baseline.txt
echo CALL1_DONE
