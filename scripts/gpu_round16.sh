#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 400 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_v5.log 2>&1; echo FULL_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/G_r50.log 2>&1; echo A=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/G_bert_packed.log 2>&1; echo B=$?
BPS_BERT_SPLIT_QKV=1 timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/G_bert_split.log 2>&1; echo C=$?
tail -2 gpurun_out/pytest_v5.log; tail -1 gpurun_out/G_r50.log; tail -1 gpurun_out/G_bert_packed.log; tail -1 gpurun_out/G_bert_split.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
