#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
rocm-smi --showclocks --showpower --showtemp > gpurun_out/f4_smi.log 2>&1 || true
grep -iE "sclk|power|temp|junction" gpurun_out/f4_smi.log | head -6
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/f4_gputests.log 2>&1
echo "gputests rc=$?"; tail -2 gpurun_out/f4_gputests.log
for i in 1 2 3; do
  timeout 600 python bench.py --steps 20 --warmup 3 --json-out gpurun_out/f4_b$i.json > gpurun_out/f4_b$i.log 2>&1
  python -c "import json;print('bench$i', round(json.load(open('gpurun_out/f4_b$i.json'))['ms_per_step'],2))" || tail -2 gpurun_out/f4_b$i.log
done
timeout 600 python bench.py --steps 300 --warmup 3 --json-out gpurun_out/f4_soak.json > gpurun_out/f4_soak.log 2>&1
python -c "import json;print('soak300', round(json.load(open('gpurun_out/f4_soak.json'))['ms_per_step'],2))" || tail -2 gpurun_out/f4_soak.log
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats -d /tmp/f4prof -o f4 -- python /root/repo/bench.py --steps 3 --warmup 1 > /dev/null 2>&1
cd /root/repo
python tools/prof_summary.py /tmp/f4prof/f4_results.db 45 gpurun_out/f4_kernels.txt > /dev/null 2>&1
head -14 gpurun_out/f4_kernels.txt
echo F4_DONE
