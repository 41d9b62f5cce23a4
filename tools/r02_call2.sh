#!/bin/bash
# In-app A/B of the v2 GEMM dispatch on the 160L headline bench.
set -x
export TMPDIR=/tmp
cd /root/repo
timeout 600 python -m pytest tests/test_ops_gpu.py -q -m gpu -k "gemm2 or hblt or linear" > gpurun_out/r02c2_tests.log 2>&1
echo "tests rc=$?"; tail -3 gpurun_out/r02c2_tests.log
for mode in "" "fwd" "dgrad,wgrad" "1"; do
  tag=${mode:-off}; tag=${tag//,/}
  SKY_GEMM2=$mode timeout 600 python bench.py --steps 10 --warmup 2 \
    --json-out gpurun_out/r02c2_b160_${tag}.json > gpurun_out/r02c2_b160_${tag}.log 2>&1
  echo "bench $tag rc=$?"
  python -c "import json;d=json.load(open('gpurun_out/r02c2_b160_${tag}.json'));print('$tag', round(d['ms_per_step'],2))" 2>/dev/null || tail -2 gpurun_out/r02c2_b160_${tag}.log
done
echo CALL2_DONE
