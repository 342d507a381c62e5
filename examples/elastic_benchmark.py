#!/usr/bin/env python3
"""Elastic suspend/resume demo (reference
example/pytorch/elastic_benchmark_byteps.py:124-133): train, suspend the
engine, resume with (potentially) new cluster topology — declared tensor
keys stay stable across the cycle."""

import torch

import byteps_amd.torch as bps
from byteps_amd.models import mnist_mlp

bps.init()
device = torch.device("cuda", bps.local_rank()) \
    if torch.cuda.is_available() else torch.device("cpu")

model = mnist_mlp().to(device)
opt = torch.optim.SGD(model.parameters(), lr=0.05)
opt = bps.DistributedOptimizer(opt, named_parameters=model.named_parameters())

def train(steps):
    for _ in range(steps):
        x = torch.randn(32, 1, 28, 28, device=device)
        y = torch.randint(0, 10, (32,), device=device)
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()

train(10)
print("rank %d: suspending" % bps.rank(), flush=True)
bps.suspend()
# ... scaling event happens here (workers added/removed, env updated) ...
bps.resume(num_workers=1, num_servers=0)
print("resumed; keys preserved", flush=True)
train(10)
bps.shutdown()
print("elastic cycle complete", flush=True)
