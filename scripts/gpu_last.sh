#!/bin/bash
set -x
for attempt in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 1000 -- 'set -x
cd /root/repo
timeout 350 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_last1.log 2>&1; echo R1=$?
timeout 350 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_last2.log 2>&1; echo R2=$?
timeout 240 python bench.py > gpurun_out/J_r50_defaults.log 2>&1; echo A=$?
tail -1 gpurun_out/pytest_last1.log; tail -1 gpurun_out/pytest_last2.log; tail -1 gpurun_out/J_r50_defaults.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
