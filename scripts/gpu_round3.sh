#!/bin/bash
# Retry GPU run 3 until a slot frees (exit 3 = busy, nothing charged).
set -x
for attempt in $(seq 1 20); do
  /usr/local/graft/bin/gpurun --timeout 1500 -- 'set -x
cd /root/repo
timeout 420 python -m pytest tests/test_fused_bn_gpu.py tests/test_ps_gpu.py -m gpu -q > gpurun_out/pytest_bn.log 2>&1; echo BN_RC=$?
timeout 420 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_bn.log 2>&1; echo R50_RC=$?
timeout 420 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert64.log 2>&1; echo BERT_RC=$?
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace -d /root/repo/gpurun_out/prof2 -- python /root/repo/bench.py --steps 10 --warmup 5 --graph off > /root/repo/gpurun_out/prof2.log 2>&1; echo PROF_RC=$?
tail -2 /root/repo/gpurun_out/pytest_bn.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then
    exit $rc
  fi
  sleep 120
done
exit 3
