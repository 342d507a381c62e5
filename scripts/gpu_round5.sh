#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 900 -- 'set -x
cd /root/repo
timeout 240 python -m pytest tests/test_fused_bn_gpu.py -m gpu -q > gpurun_out/pytest_bn3.log 2>&1; echo BN_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_bn3.log 2>&1; echo R50_RC=$?
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof4 -- python /root/repo/bench.py --model bert-large --steps 8 --warmup 4 --graph off > /root/repo/gpurun_out/prof4_bert.log 2>&1; echo PROF_RC=$?
tail -2 /root/repo/gpurun_out/pytest_bn3.log; tail -1 /root/repo/gpurun_out/bench_r50_bn3.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
