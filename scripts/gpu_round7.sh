#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 500 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu_full.log 2>&1; echo FULL_RC=$?
timeout 200 python scripts/micro_ln_sdpa.py > gpurun_out/micro_ln_sdpa.log 2>&1; echo MICRO_RC=$?
BPS_SDPA_BACKEND=efficient timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert_eff.log 2>&1; echo BERT_EFF_RC=$?
timeout 300 python bench.py --steps 20 --warmup 8 --compression onebit > gpurun_out/bench_r50_onebit.log 2>&1; echo ONEBIT_RC=$?
tail -2 /root/repo/gpurun_out/pytest_gpu_full.log; cat /root/repo/gpurun_out/micro_ln_sdpa.log; tail -1 /root/repo/gpurun_out/bench_bert_eff.log; tail -1 /root/repo/gpurun_out/bench_r50_onebit.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
