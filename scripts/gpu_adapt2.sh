#!/bin/bash
set -x
for attempt in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 700 -- 'set -x
cd /root/repo
timeout 280 python -m pytest tests/test_fused_bn_gpu.py tests/test_fused_ln_gpu.py -m gpu -q > gpurun_out/pytest_adapt2.log 2>&1; echo T=$?
timeout 150 python scripts/micro_bn_ln.py > gpurun_out/micro_adapt2.log 2>&1; echo M=$?
timeout 220 python bench.py --steps 30 --warmup 10 > gpurun_out/M_r50.log 2>&1; echo A=$?
tail -1 gpurun_out/pytest_adapt2.log; grep -v amdgpu gpurun_out/micro_adapt2.log | head -14; tail -1 gpurun_out/M_r50.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
