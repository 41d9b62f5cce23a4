#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/final_gputests.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/final_gputests.log
timeout 600 python bench.py --steps 25 --warmup 3 --json-out gpurun_out/final_b160.json > gpurun_out/final_b160.log 2>&1
python -c "import json;print('bench160', round(json.load(open('gpurun_out/final_b160.json'))['ms_per_step'],2))"
timeout 600 python bench.py --steps 60 --warmup 3 --json-out gpurun_out/final_soak.json > gpurun_out/final_soak.log 2>&1
python -c "import json;print('soak60  ', round(json.load(open('gpurun_out/final_soak.json'))['ms_per_step'],2))"
timeout 420 python bench.py --layers 24 --steps 15 --warmup 3 --json-out gpurun_out/final_b24.json > gpurun_out/final_b24.log 2>&1
python -c "import json;print('bench24 ', round(json.load(open('gpurun_out/final_b24.json'))['ms_per_step'],2))"
timeout 420 python bench.py --layers 24 --dtype fp32 --steps 10 --warmup 3 --json-out gpurun_out/final_b24fp32.json > gpurun_out/final_b24fp32.log 2>&1
python -c "import json;print('bench24fp32', round(json.load(open('gpurun_out/final_b24fp32.json'))['ms_per_step'],2))"
cd /tmp
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/fprof -o f -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/final_prof.log 2>&1
cd /root/repo
DB=$(ls /tmp/fprof/*.db | head -1)
python tools/prof_summary.py "$DB" 45 gpurun_out/final_kernels.txt > /dev/null 2>&1
head -20 gpurun_out/final_kernels.txt
echo FINAL_DONE
