#!/bin/bash
set -x
for attempt in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 800 -- 'set -x
cd /root/repo
timeout 300 python -m pytest tests/test_fused_bn_gpu.py tests/test_fused_ln_gpu.py -m gpu -q > gpurun_out/pytest_adapt.log 2>&1; echo T=$?
timeout 150 python scripts/micro_bn_ln.py > gpurun_out/micro_adapt.log 2>&1; echo M=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/L_r50.log 2>&1; echo A=$?
timeout 280 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/L_bert.log 2>&1; echo B=$?
tail -1 gpurun_out/pytest_adapt.log; grep -v amdgpu gpurun_out/micro_adapt.log | head -14; tail -1 gpurun_out/L_r50.log; tail -1 gpurun_out/L_bert.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
