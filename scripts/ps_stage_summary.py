#!/usr/bin/env python3
"""Summarize a byteps_amd chrome trace (BPS_TRACE_ON=1 output): per-stage
total/mean wall time per step, and per-step aggregate span — the tool for
finding where the PS path's per-step overhead lives.

Usage: python scripts/ps_stage_summary.py <trace_dir>/<rank>/comm.json
"""

import json
import sys
from collections import defaultdict


def main(path: str) -> None:
    with open(path) as f:
        data = json.load(f)
    ev = data["traceEvents"]
    if not ev:
        print("no events")
        return
    by_stage = defaultdict(list)          # stage -> [dur_us]
    by_step_stage = defaultdict(lambda: defaultdict(list))
    steps = set()
    for e in ev:
        st = e["args"]["step"]
        steps.add(st)
        by_stage[e["name"]].append(e["dur"])
        by_step_stage[st][e["name"]].append(e["dur"])
    nsteps = len(steps)
    print("%d events over %d steps" % (len(ev), nsteps))
    print("%-18s %8s %10s %10s %10s" %
          ("stage", "count", "mean_us", "p50_us", "sum_ms/step"))
    for name, durs in sorted(by_stage.items()):
        durs.sort()
        print("%-18s %8d %10.0f %10.0f %10.2f" %
              (name, len(durs), sum(durs) / len(durs),
               durs[len(durs) // 2], sum(durs) / 1000.0 / max(1, nsteps)))
    # per-step critical span: last end - first begin across all keys
    spans = []
    for st in sorted(steps):
        begin = min(e["ts"] for e in ev if e["args"]["step"] == st)
        end = max(e["ts"] + e["dur"] for e in ev if e["args"]["step"] == st)
        spans.append((end - begin) / 1000.0)
    spans.sort()
    print("comm span per step: p50 %.2f ms  max %.2f ms"
          % (spans[len(spans) // 2], spans[-1]))


if __name__ == "__main__":
    main(sys.argv[1])
