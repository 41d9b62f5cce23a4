#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/f2_gputests.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/f2_gputests.log
run() {
  timeout 600 env $2 python bench.py --steps $3 --warmup 3 --json-out gpurun_out/f2_$1.json > gpurun_out/f2_$1.log 2>&1
  python -c "import json;d=json.load(open('gpurun_out/f2_$1.json'));print('$1', round(d['ms_per_step'],2))" || tail -2 gpurun_out/f2_$1.log
}
run b160 "" 25
run soak "" 100
run sgd8 "SKY_SGD_UNROLL=8" 15
run qkv2 "SKY_GEMM2=fwd SKY_GEMM2_SHAPES=4096x3072x1024,4096x4096x1024" 15
run fusedbwd "SKY_ATTN_FUSED_BWD=1" 15
cd /tmp
timeout 400 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT -d /tmp/trvp -o trv -- python /root/repo/tools/k3_pmc.py > /dev/null 2>&1
cd /root/repo
python tools/pmc_summary.py /tmp/trvp/trv_results.db 2>/dev/null | grep -A6 "attn_fwd_kernel" | head -8
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /tmp/f2prof -o f2 -- python /root/repo/bench.py --steps 3 --warmup 1 > /dev/null 2>&1
cd /root/repo
python tools/prof_summary.py /tmp/f2prof/f2_results.db 45 gpurun_out/f2_kernels.txt > /dev/null 2>&1
head -16 gpurun_out/f2_kernels.txt
echo F2_DONE
