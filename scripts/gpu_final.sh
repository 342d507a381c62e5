#!/bin/bash
set -x
for attempt in $(seq 1 30); do
  /usr/local/graft/bin/gpurun --timeout 1500 -- 'set -x
cd /root/repo
timeout 500 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu_final.log 2>&1; echo FULL_RC=$?
timeout 200 python -c "import __graft_entry__; __graft_entry__.smoke()" > gpurun_out/smoke_final.log 2>&1; echo SMOKE_RC=$?
timeout 240 python bench.py --steps 30 --warmup 10 > gpurun_out/F_r50.log 2>&1; echo A=$?
timeout 300 python bench.py --model bert-large --steps 20 --warmup 8 > gpurun_out/F_bert.log 2>&1; echo B=$?
timeout 300 python bench.py --model vgg16 --steps 20 --warmup 8 > gpurun_out/F_vgg.log 2>&1; echo C=$?
timeout 240 python bench.py --model mlp --steps 50 --warmup 10 > gpurun_out/F_mlp.log 2>&1; echo D=$?
timeout 300 python bench.py --steps 20 --warmup 5 --compression onebit > gpurun_out/F_r50_onebit.log 2>&1; echo E=$?
BPS_FORCE_DISTRIBUTED=1 BPS_NUM_SERVER=1 BPS_SERVER_URIS=127.0.0.1:29637 timeout 300 bash -c "python -m byteps_amd.server & SRV=\$!; BPS_SERVER_PORT=29637 python bench.py --steps 20 --warmup 5 --graph off > gpurun_out/F_r50_ps_raw.log 2>&1; kill \$SRV" 2>/dev/null; echo F=$?
tail -2 gpurun_out/pytest_gpu_final.log
for f in gpurun_out/F_*.log; do echo "== $f"; tail -1 $f; done'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 120
done
exit 3
