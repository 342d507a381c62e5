"""In-tree build of the native core: one hipcc invocation cross-compiles
the gfx950 kernels + host code into ``byteps_amd/ops/_core.so`` (the .so
travels with the repo snapshot to GPU boxes; no JIT cache involved)."""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT = os.path.join(HERE, "_core.so")

SOURCES = ["core.cc", "kv.cc", "server.cc", "cpu_reducer.cc", "rdma.cc", "blaslt.cc",
           "kernels.hip", "compress.hip", "bn.hip", "ln.hip"]


def _newer_than_out(paths) -> bool:
    if not os.path.exists(OUT):
        return True
    out_mtime = os.path.getmtime(OUT)
    return any(os.path.getmtime(p) > out_mtime for p in paths)


def build(force: bool = False, arch: str = "gfx950", verbose: bool = True) -> str:
    import pybind11
    srcs = [os.path.join(CSRC, s) for s in SOURCES]
    hdrs = [os.path.join(CSRC, h) for h in os.listdir(CSRC)
            if h.endswith(".h")]
    if not force and not _newer_than_out(srcs + hdrs):
        return OUT
    hipcc = os.environ.get("HIPCC", "hipcc")
    tmp_out = OUT + ".tmp"
    cmd = [
        hipcc, "--offload-arch=" + arch, "-O3", "-std=c++17", "-fPIC",
        "-shared", "-fopenmp", "-pthread", "-fvisibility=hidden",
        "-Wno-unused-result",
        "-L/opt/rocm/lib", "-lhipblaslt",
        "-I", pybind11.get_include(),
        "-I", sysconfig.get_paths()["include"],
        *srcs,
        "-o", tmp_out,
    ]
    if verbose:
        print("[byteps_amd build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    # atomic swap: a concurrently-spawning process must never import a
    # half-written .so
    os.replace(tmp_out, OUT)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
