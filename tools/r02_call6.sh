#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
timeout 300 python -m pytest tests/test_ops_gpu.py -q -m gpu -k "layernorm or gelu" > gpurun_out/c6_tests.log 2>&1
echo "tests rc=$?"; tail -2 gpurun_out/c6_tests.log
run() {
  timeout 420 env $2 python bench.py --steps 15 --warmup 2 --json-out gpurun_out/c6_$1.json > gpurun_out/c6_$1.log 2>&1
  python -c "import json;d=json.load(open('gpurun_out/c6_$1.json'));print('$1', round(d['ms_per_step'],2))" || tail -2 gpurun_out/c6_$1.log
}
run oldln "SKY_LN_SPLIT_WB=1"
run g512 ""
run g768 "SKY_LN_CS_GRID=768"
run g1024 "SKY_LN_CS_GRID=1024"
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/c6_prof -o c6 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/c6_prof.log 2>&1
cd /root/repo
DB=$(ls gpurun_out/c6_prof/*.db | head -1)
python tools/prof_summary.py "$DB" 45 gpurun_out/c6_kernels.txt > /dev/null 2>&1
grep -E "ln_|gelu|colsum|gemm2" gpurun_out/c6_kernels.txt | head -12
echo CALL6_DONE
