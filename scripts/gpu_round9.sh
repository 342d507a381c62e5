#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 500 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu_full2.log 2>&1; echo FULL_RC=$?
timeout 200 python scripts/micro_ln_sdpa.py > gpurun_out/micro2.log 2>&1; echo MICRO_RC=$?
timeout 300 python bench.py --steps 20 --warmup 5 --compression onebit > gpurun_out/bench_r50_onebit2.log 2>&1; echo ONEBIT_RC=$?
timeout 300 python bench.py --steps 20 --warmup 5 --compression topk > gpurun_out/bench_r50_topk.log 2>&1; echo TOPK_RC=$?
tail -2 gpurun_out/pytest_gpu_full2.log; grep -v amdgpu gpurun_out/micro2.log; tail -1 gpurun_out/bench_r50_onebit2.log; tail -1 gpurun_out/bench_r50_topk.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
