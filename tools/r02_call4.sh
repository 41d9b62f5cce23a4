#!/bin/bash
# LN/GELU bwd fusion A/B: kernel-level correctness (pytest subset) + in-app step time.
set -x
export TMPDIR=/tmp
cd /root/repo
timeout 600 python -m pytest tests/test_ops_gpu.py -q -m gpu -k "layernorm or gelu or gemm2" > gpurun_out/r02c4_tests.log 2>&1
echo "tests rc=$?"; tail -3 gpurun_out/r02c4_tests.log
for mode in new old; do
  if [ $mode = old ]; then export SKY_LN_SPLIT_WB=1 SKY_GELU_SPLIT_DB=1; else unset SKY_LN_SPLIT_WB SKY_GELU_SPLIT_DB; fi
  timeout 420 python bench.py --steps 15 --warmup 2 --json-out gpurun_out/r02c4_${mode}.json > gpurun_out/r02c4_${mode}.log 2>&1
  python -c "import json;d=json.load(open('gpurun_out/r02c4_${mode}.json'));print('$mode', round(d['ms_per_step'],2))" || tail -2 gpurun_out/r02c4_${mode}.log
done
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r02c4_prof -o c4 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/r02c4_prof.log 2>&1
cd /root/repo
DB=$(ls gpurun_out/r02c4_prof/*.db | head -1)
python tools/prof_summary.py "$DB" 40 gpurun_out/r02c4_kernels.txt > /dev/null 2>&1
grep -E "ln_|gelu|colsum" gpurun_out/r02c4_kernels.txt | head -12
echo CALL4_DONE
