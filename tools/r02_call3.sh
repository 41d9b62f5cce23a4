#!/bin/bash
# Per-site in-app A/B of v2 GEMM dispatch: one bench run per (site, shape).
set -x
export TMPDIR=/tmp
cd /root/repo
run() {
  tag=$1; site=$2; shapes=$3
  SKY_GEMM2=$site SKY_GEMM2_SHAPES=$shapes timeout 420 python bench.py --steps 15 --warmup 2 \
    --json-out gpurun_out/ab_${tag}.json > gpurun_out/ab_${tag}.log 2>&1
  python -c "import json;d=json.load(open('gpurun_out/ab_${tag}.json'));print('${tag}', round(d['ms_per_step'],2))" || tail -2 gpurun_out/ab_${tag}.log
}
run off0 "" ""
run fqkv fwd 4096x3072x1024
run fproj fwd 4096x1024x1024
run fup fwd 4096x4096x1024
run fdn fwd 4096x1024x4096
run dqkv dgrad 4096x1024x3072
run dproj dgrad 4096x1024x1024
run dup dgrad 4096x1024x4096
run ddn dgrad 4096x4096x1024
run wqkv wgrad 3072x1024x4096
run wproj wgrad 1024x1024x4096
run wup wgrad 4096x1024x4096
run wdn wgrad 1024x4096x4096
run off1 "" ""
echo CALL3_DONE
