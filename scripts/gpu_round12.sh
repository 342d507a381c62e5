#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1500 -- 'set -x
cd /root/repo
timeout 500 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu_v3.log 2>&1; echo FULL_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_r50_final.log 2>&1; echo R50_RC=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/bench_bert_final.log 2>&1; echo BERT_RC=$?
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace -d /tmp/prof5 -- python /root/repo/bench.py --model bert-large --steps 6 --warmup 3 --graph off > /root/repo/gpurun_out/prof5.log 2>&1; echo PROF_BERT_RC=$?
python /root/repo/scripts/prof_summary.py /tmp/prof5 250 /root/repo/gpurun_out/prof_bert_summary.txt > /dev/null 2>&1; echo SUM1_RC=$?
timeout 300 rocprofv3 --kernel-trace -d /tmp/prof6 -- python /root/repo/bench.py --steps 10 --warmup 5 --graph off > /root/repo/gpurun_out/prof6.log 2>&1; echo PROF_R50_RC=$?
python /root/repo/scripts/prof_summary.py /tmp/prof6 150 /root/repo/gpurun_out/prof_r50_summary.txt > /dev/null 2>&1; echo SUM2_RC=$?
cd /root/repo
tail -2 gpurun_out/pytest_gpu_v3.log; tail -1 gpurun_out/bench_r50_final.log; tail -1 gpurun_out/bench_bert_final.log; head -14 gpurun_out/prof_bert_summary.txt'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
