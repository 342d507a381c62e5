"""PS path on a real GPU: worker on cuda:0, native C++ server on the same
box (CPU) — exercises HIP codec kernels + pinned D2H/H2D staging + TCP KV
end-to-end (BASELINE.json configs 3-5 in miniature)."""

import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

_WORKER = r"""
import os, sys, torch
import byteps_amd.torch as bps

mode = sys.argv[1]
bps.init()
torch.manual_seed(0)
m = torch.nn.Sequential(
    torch.nn.Linear(64, 128), torch.nn.ReLU(),
    torch.nn.Linear(128, 10)).cuda()
opt = torch.optim.SGD(m.parameters(), lr=0.05)
params = {}
if mode == "onebit":
    params = {"compressor_type": "onebit", "ef_type": "vanilla"}
elif mode == "topk_full":
    params = {"compressor_type": "topk", "compressor_k": 1 << 20}
elif mode == "dithering":
    params = {"compressor_type": "dithering", "compressor_k": 64}
opt = bps.DistributedOptimizer(
    opt, named_parameters=m.named_parameters(),
    compression_params=params or None)

torch.manual_seed(1)
x = torch.randn(32, 64).cuda()
y = torch.randint(0, 10, (32,)).cuda()
losses = []
for i in range(15):
    opt.zero_grad()
    loss = torch.nn.functional.cross_entropy(m(x), y)
    loss.backward()
    losses.append(float(loss))
    opt.step()
torch.cuda.synchronize()
assert losses[-1] < losses[0], losses
if mode == "raw":
    # raw path must match a local (uncompressed, world=1) run exactly
    torch.manual_seed(0)
    m2 = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 10)).cuda()
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.05)
    for i in range(15):
        opt2.zero_grad()
        torch.nn.functional.cross_entropy(m2(x), y).backward()
        opt2.step()
    for p1, p2 in zip(m.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=1e-4, atol=1e-5), mode
print("WORKER_OK", mode, losses[0], losses[-1])
bps.shutdown()
"""


@pytest.mark.parametrize("mode", ["raw", "onebit", "topk_full", "dithering"])
def test_ps_gpu_worker(mode, tmp_path):
    from byteps_amd.ops import _core
    srv = _core.Server(0, 2, False)
    srv.start()
    try:
        script = tmp_path / "worker.py"
        script.write_text(_WORKER)
        repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env = dict(os.environ)
        env.update({
            "BPS_FORCE_DISTRIBUTED": "1",
            "BPS_NUM_SERVER": "1",
            "BPS_SERVER_URIS": "127.0.0.1:%d" % srv.port,
            "BPS_MIN_COMPRESS_BYTES": "0",
            "PYTHONPATH": repo_root + os.pathsep + os.environ.get(
                "PYTHONPATH", ""),
        })
        out = subprocess.run(
            [sys.executable, str(script), mode], env=env,
            capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        assert "WORKER_OK" in out.stdout
    finally:
        srv.stop()
