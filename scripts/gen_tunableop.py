"""Generate a PyTorch TunableOp result file for the BERT-large GEMM
shapes on gfx950 (hipBLASLt algorithm selection).  Run on a GPU box:

    python scripts/gen_tunableop.py

Writes gpurun_out/tunableop_gfx950.csv — commit it to
byteps_amd/tuning/ so bench.py can load it (tuning itself is too slow
to run inside a timed benchmark)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "20")
os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "30")

import torch  # noqa: E402

out = os.path.join("gpurun_out", "tunableop_gfx950.csv")
os.makedirs("gpurun_out", exist_ok=True)
torch.cuda.tunable.enable(True)
torch.cuda.tunable.tuning_enable(True)
torch.cuda.tunable.set_filename(out)

from byteps_amd.models import bert_large  # noqa: E402

m = bert_large().cuda()
opt = torch.optim.SGD(m.parameters(), lr=0.01)
ids = torch.randint(0, 30522, (64, 128), device="cuda")
lab = torch.randint(0, 30522, (64, 128), device="cuda")
for i in range(3):
    opt.zero_grad()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = m.loss(ids, lab)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    print("step", i, "loss", float(loss), flush=True)

# results stream to the file during tuning; flush explicitly when available
getattr(torch.cuda.tunable, "write_file", lambda: None)()
print("wrote", out, "entries:", len(open(out).readlines()))
