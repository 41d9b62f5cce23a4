#!/bin/bash
# Clean A/B: (a) baseline all-off, (b) fwd-gelu-fused default, (c) +LN/gelu bwd fusion
set -x
export TMPDIR=/tmp
cd /root/repo
run() {
  tag=$1
  timeout 420 env $2 python bench.py --steps 15 --warmup 2 \
    --json-out gpurun_out/c5_${tag}.json > gpurun_out/c5_${tag}.log 2>&1
  python -c "import json;d=json.load(open('gpurun_out/c5_${tag}.json'));print('${tag}', round(d['ms_per_step'],2))" || tail -2 gpurun_out/c5_${tag}.log
}
run all_old "SKY_GEMM2=0 SKY_LN_SPLIT_WB=1 SKY_GELU_SPLIT_DB=1"
run g2fwd   "SKY_LN_SPLIT_WB=1 SKY_GELU_SPLIT_DB=1"
run lnfuse  "SKY_GELU_SPLIT_DB=1"
run gelufuse "SKY_LN_SPLIT_WB=1"
run allnew  ""
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/c5_prof -o c5 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/c5_prof.log 2>&1
cd /root/repo
DB=$(ls gpurun_out/c5_prof/*.db | head -1)
python tools/prof_summary.py "$DB" 45 gpurun_out/c5_kernels.txt > /dev/null 2>&1
grep -E "ln_|gelu|colsum|gemm2" gpurun_out/c5_kernels.txt | head -14
echo CALL5_DONE
