#!/usr/bin/env python3
"""Synthetic-data training benchmark (reference vehicle:
example/pytorch/benchmark_byteps.py — img/sec mean ± 1.96 σ over iters).

  python -m torch.distributed.run --nproc-per-node 8 --master-addr 127.0.0.1 \
      examples/benchmark_byteps_amd.py --model resnet50 --num-iters 10
"""

import argparse
import time

import numpy as np
import torch

import byteps_amd.torch as bps
from byteps_amd import models
from byteps_amd.torch.parallel import DistributedDataParallel as DDP

p = argparse.ArgumentParser()
p.add_argument("--model", default="resnet50",
               choices=["resnet50", "resnet101", "vgg16"])
p.add_argument("--batch-size", type=int, default=64)
p.add_argument("--num-warmup-batches", type=int, default=10)
p.add_argument("--num-batches-per-iter", type=int, default=10)
p.add_argument("--num-iters", type=int, default=10)
p.add_argument("--no-cuda", action="store_true")
p.add_argument("--compression", default=None,
               choices=[None, "onebit", "topk", "randomk", "dithering"])
args = p.parse_args()

bps.init()
use_cuda = not args.no_cuda and torch.cuda.is_available()
device = torch.device("cuda", bps.local_rank()) if use_cuda else "cpu"

model = getattr(models, args.model)().to(device)
if use_cuda:
    model = model.to(memory_format=torch.channels_last)
cparams = {"compressor_type": args.compression,
           "compressor_k": 1000} if args.compression else None
net = DDP(model, broadcast_buffers=False, compression_params=cparams)
opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)

data = torch.randn(args.batch_size, 3, 224, 224, device=device)
if use_cuda:
    data = data.to(memory_format=torch.channels_last)
target = torch.randint(0, 1000, (args.batch_size,), device=device)
loss_fn = torch.nn.CrossEntropyLoss()


def benchmark_step():
    net.zero_grad_buckets()
    if use_cuda:
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = loss_fn(net(data), target)
    else:
        loss = loss_fn(net(data), target)
    loss.backward()
    opt.step()


def log(s):
    if bps.rank() == 0:
        print(s, flush=True)


log("Model: %s, batch %d per %s" % (args.model, args.batch_size, device))
for _ in range(args.num_warmup_batches):
    benchmark_step()

img_secs = []
for i in range(args.num_iters):
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.num_batches_per_iter):
        benchmark_step()
    if use_cuda:
        torch.cuda.synchronize()
    img_sec = args.batch_size * args.num_batches_per_iter / (time.time() - t0)
    log("Iter #%d: %.1f img/sec per rank" % (i, img_sec))
    img_secs.append(img_sec)

img_sec_mean = np.mean(img_secs)
img_sec_conf = 1.96 * np.std(img_secs)
log("Img/sec per rank: %.1f +- %.1f" % (img_sec_mean, img_sec_conf))
log("Total img/sec on %d rank(s): %.1f +- %.1f" %
    (bps.size(), bps.size() * img_sec_mean, bps.size() * img_sec_conf))
bps.shutdown()
