#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 900 -- 'set -x
cd /root/repo
timeout 300 python -m pytest tests/test_fused_bn_gpu.py tests/test_gpu_kernels.py -m gpu -q > gpurun_out/pytest_bn4.log 2>&1; echo BN_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_mask.log 2>&1; echo R50_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_mask2.log 2>&1; echo R50B_RC=$?
tail -2 gpurun_out/pytest_bn4.log; tail -1 gpurun_out/bench_r50_mask.log; tail -1 gpurun_out/bench_r50_mask2.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
