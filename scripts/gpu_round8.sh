#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 200 python scripts/micro_ln_sdpa.py > gpurun_out/micro_ln_sdpa.log 2>&1; echo MICRO_RC=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert_v3.log 2>&1; echo BERT_RC=$?
BPS_TRACE_ON=1 BPS_TRACE_DIR=gpurun_out/traces BPS_TRACE_START_STEP=2 BPS_TRACE_END_STEP=6 timeout 300 python bench.py --steps 8 --warmup 2 --compression onebit > gpurun_out/bench_r50_onebit_trace.log 2>&1; echo ONEBIT_RC=$?
cat gpurun_out/micro_ln_sdpa.log; tail -1 gpurun_out/bench_bert_v3.log; tail -1 gpurun_out/bench_r50_onebit_trace.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
