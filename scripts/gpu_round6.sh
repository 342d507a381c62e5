#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 300 python -m pytest tests/test_fused_bn_gpu.py tests/test_fused_ln_gpu.py -m gpu -q > gpurun_out/pytest_bnln.log 2>&1; echo BNLN_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_v3.log 2>&1; echo R50_RC=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert_v2.log 2>&1; echo BERT_RC=$?
timeout 300 python bench.py --model vgg16 --steps 20 --warmup 8 > gpurun_out/bench_vgg_v2.log 2>&1; echo VGG_RC=$?
tail -2 /root/repo/gpurun_out/pytest_bnln.log; tail -1 /root/repo/gpurun_out/bench_r50_v3.log; tail -1 /root/repo/gpurun_out/bench_bert_v2.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
