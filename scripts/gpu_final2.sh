#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 400 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_final.log 2>&1; echo FULL_RC=$?
timeout 200 python -c "import __graft_entry__; __graft_entry__.smoke()" > gpurun_out/smoke_final2.log 2>&1; echo SMOKE_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/I_r50.log 2>&1; echo A=$?
timeout 240 python bench.py --steps 30 --warmup 10 --dtype fp32 > gpurun_out/I_r50_fp32.log 2>&1; echo B=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/I_bert.log 2>&1; echo C=$?
tail -2 gpurun_out/pytest_final.log; tail -1 gpurun_out/smoke_final2.log; tail -1 gpurun_out/I_r50.log; tail -1 gpurun_out/I_r50_fp32.log; tail -1 gpurun_out/I_bert.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
