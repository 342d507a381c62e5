#!/usr/bin/env python3
"""MNIST-shaped MLP end to end on CPU/gloo or GPU/RCCL — the minimal
plumbing example (BASELINE.json config 1).  Synthetic data (no network
access for datasets).

  python -m torch.distributed.run --nproc-per-node 2 --master-addr 127.0.0.1 \
      examples/train_mnist_mlp.py
"""

import torch
import torch.nn.functional as F

import byteps_amd.torch as bps
from byteps_amd.models import mnist_mlp

bps.init()
device = torch.device("cuda", bps.local_rank()) \
    if torch.cuda.is_available() else torch.device("cpu")

torch.manual_seed(1234)
model = mnist_mlp().to(device)
opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
opt = bps.DistributedOptimizer(opt, named_parameters=model.named_parameters())
bps.broadcast_parameters(model.state_dict(), root_rank=0)
bps.broadcast_optimizer_state(opt, root_rank=0)

torch.manual_seed(bps.rank())        # different data per rank
for step in range(50):
    x = torch.randn(64, 1, 28, 28, device=device)
    y = torch.randint(0, 10, (64,), device=device)
    opt.zero_grad()
    loss = F.cross_entropy(model(x), y)
    loss.backward()
    opt.step()
    if step % 10 == 0 and bps.rank() == 0:
        print("step %d loss %.4f" % (step, loss.item()), flush=True)

bps.shutdown()
