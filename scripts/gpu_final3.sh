#!/bin/bash
set -x
for attempt in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 700 -- 'set -x
cd /root/repo
timeout 350 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_final3.log 2>&1; echo T=$?
timeout 220 python bench.py --steps 30 --warmup 10 > gpurun_out/N_r50.log 2>&1; echo A=$?
timeout 280 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/N_bert.log 2>&1; echo B=$?
tail -1 gpurun_out/pytest_final3.log; tail -1 gpurun_out/N_r50.log; tail -1 gpurun_out/N_bert.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
