#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 900 -- 'set -x
cd /root/repo
timeout 400 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_v6.log 2>&1; echo FULL_RC=$?
timeout 200 python -c "import __graft_entry__; __graft_entry__.smoke()" > gpurun_out/smoke_v6.log 2>&1; echo SMOKE_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/H_r50.log 2>&1; echo A=$?
tail -2 gpurun_out/pytest_v6.log; tail -1 gpurun_out/smoke_v6.log; tail -1 gpurun_out/H_r50.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
