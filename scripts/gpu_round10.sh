#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1200 -- 'set -x
cd /root/repo
timeout 600 python scripts/gen_tunableop.py > gpurun_out/gen_tunable.log 2>&1; echo GEN_RC=$?
tail -3 gpurun_out/gen_tunable.log
# quick A/B with the fresh file in place
mkdir -p byteps_amd/tuning && cp gpurun_out/tunableop_gfx950.csv byteps_amd/tuning/ 2>/dev/null
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert_tuned.log 2>&1; echo BERT_TUNED_RC=$?
tail -1 gpurun_out/bench_bert_tuned.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
