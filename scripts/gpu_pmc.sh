#!/bin/bash
set -x
for attempt in $(seq 1 40); do
  /usr/local/graft/bin/gpurun --timeout 800 -- 'set -x
cd /tmp && export TMPDIR=/tmp
timeout 300 python /root/repo/scripts/micro_bn_ln.py > /root/repo/gpurun_out/micro_bnln.log 2>&1; echo TIME_RC=$?
timeout 300 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT,SQ_WAVES -d /tmp/pmc1 -- python /root/repo/scripts/micro_bn_ln.py > /root/repo/gpurun_out/pmc1.log 2>&1; echo PMC_RC=$?
cd /tmp/pmc1 2>/dev/null && ls -la && find . -name "*.db" -o -name "*.csv" | head
python - <<PYEOF > /root/repo/gpurun_out/pmc_summary.txt 2>&1
import sqlite3, glob, collections
paths = sorted(glob.glob("/tmp/pmc1/**/*.db", recursive=True))
print("dbs:", paths)
if paths:
    db = sqlite3.connect(paths[-1]); cur = db.cursor()
    tables = [r[0] for r in cur.execute("SELECT name FROM sqlite_master WHERE type=\"table\"")]
    pmct = [t for t in tables if "pmc_event" in t]
    dispt = [t for t in tables if "kernel_dispatch" in t]
    symt = [t for t in tables if "kernel_symbol" in t]
    print("tables:", pmct, dispt, symt)
    if pmct and dispt and symt:
        u = pmct[0].replace("rocpd_pmc_event_","")
        cols = [r[1] for r in cur.execute(f"PRAGMA table_info({pmct[0]})")]
        print("pmc cols:", cols)
        for row in cur.execute(f"SELECT * FROM {pmct[0]} LIMIT 5"):
            print(row)
PYEOF
tail -12 /root/repo/gpurun_out/micro_bnln.log'
  rc=$?
  echo "gpurun attempt $attempt rc=$rc"
  if [ "$rc" != "3" ]; then exit $rc; fi
  sleep 150
done
exit 3
