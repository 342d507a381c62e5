#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 900 -- 'set -x
cd /root/repo
timeout 400 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_fp8.log 2>&1; echo FULL_RC=$?
timeout 300 python bench.py --steps 20 --warmup 5 --compression fp8 > gpurun_out/F_r50_fp8.log 2>&1; echo FP8_RC=$?
tail -2 gpurun_out/pytest_fp8.log; tail -1 gpurun_out/F_r50_fp8.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
