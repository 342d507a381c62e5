#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 400 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_v4.log 2>&1; echo FULL_RC=$?
timeout 300 python bench.py --steps 20 --warmup 5 --compression fp8 > gpurun_out/F_r50_fp8b.log 2>&1; echo FP8_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 --batch-size 128 > gpurun_out/F_r50_b128.log 2>&1; echo B128_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 --batch-size 256 > gpurun_out/F_r50_b256.log 2>&1; echo B256_RC=$?
timeout 300 python bench.py --model bert-large --batch-size 16 --seq-len 512 --steps 20 --warmup 8 > gpurun_out/F_bert_s512.log 2>&1; echo BS512_RC=$?
tail -2 gpurun_out/pytest_v4.log; for f in gpurun_out/F_r50_fp8b.log gpurun_out/F_r50_b128.log gpurun_out/F_r50_b256.log gpurun_out/F_bert_s512.log; do tail -1 $f; done'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
