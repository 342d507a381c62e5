#!/bin/bash
set -x
for attempt in $(seq 1 20); do
  /usr/local/graft/bin/gpurun --timeout 480 -- 'set -x
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace -d /tmp/proff -- python /root/repo/bench.py --steps 10 --warmup 5 --graph off > /root/repo/gpurun_out/proff.log 2>&1; echo P=$?
python /root/repo/scripts/prof_summary.py /tmp/proff 150 /root/repo/gpurun_out/prof_r50_final2.txt > /dev/null 2>&1; echo S=$?
head -16 /root/repo/gpurun_out/prof_r50_final2.txt'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
