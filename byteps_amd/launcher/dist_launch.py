#!/usr/bin/env python3
"""Multi-node fan-out launcher (reference launcher/dist_launcher.py:78-118):
reads a hostfile, SSHes the current BPS_*/DMLC_* environment and the
command to every worker host, assigning BPS_WORKER_ID by position.

  python -m byteps_amd.launcher.dist_launch --hostfile hosts \
      --env BPS_NUM_SERVER=2 -- bpslaunch python3 train.py

Hostfile: one ``host[:slots]`` per line; '#' comments.  Servers/scheduler
are launched the same way with ``--role server|scheduler``.
"""

from __future__ import annotations

import argparse
import os
import shlex
import subprocess
import sys
import threading
from typing import List

_FORWARD_PREFIXES = ("BPS_", "BYTEPS_", "DMLC_", "MASTER_", "HSA_",
                     "HIP_", "NCCL_", "RCCL_", "GLOO_")


def read_hostfile(path: str) -> List[str]:
    hosts = []
    with open(path) as f:
        for line in f:
            line = line.split("#", 1)[0].strip()
            if line:
                hosts.append(line.split(":")[0])
    return hosts


def forwarded_env(extra: List[str]) -> List[str]:
    pairs = ["%s=%s" % (k, v) for k, v in os.environ.items()
             if k.startswith(_FORWARD_PREFIXES)]
    pairs += extra
    return pairs


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--hostfile", required=True)
    p.add_argument("--role", default="worker",
                   choices=["worker", "server", "scheduler"])
    p.add_argument("--env", action="append", default=[],
                   help="extra KEY=VALUE to forward")
    p.add_argument("--ssh-opts", default="-o StrictHostKeyChecking=no")
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    args = p.parse_args()
    cmd = args.cmd[1:] if args.cmd and args.cmd[0] == "--" else args.cmd
    if not cmd:
        print("no command given", file=sys.stderr)
        return 2

    hosts = read_hostfile(args.hostfile)
    base_env = forwarded_env(args.env)
    procs = []
    outputs = {}

    def run(idx: int, host: str) -> None:
        env = list(base_env)
        env.append("BPS_ROLE=%s" % args.role)
        if args.role == "worker":
            env.append("BPS_WORKER_ID=%d" % idx)
            env.append("BPS_NUM_WORKER=%d" % len(hosts))
        remote = "env %s %s" % (
            " ".join(shlex.quote(e) for e in env),
            " ".join(shlex.quote(c) for c in cmd))
        full = ["ssh"] + shlex.split(args.ssh_opts) + [host, remote]
        proc = subprocess.Popen(full, stdout=subprocess.PIPE,
                                stderr=subprocess.STDOUT, text=True)
        procs.append(proc)
        out, _ = proc.communicate()
        outputs[host] = out

    threads = [threading.Thread(target=run, args=(i, h))
               for i, h in enumerate(hosts)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    rc = 0
    for host, out in outputs.items():
        sys.stdout.write("===== %s =====\n%s\n" % (host, out))
    for proc in procs:
        rc = rc or proc.returncode
    return rc


if __name__ == "__main__":
    sys.exit(main())
