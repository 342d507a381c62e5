#!/usr/bin/env python3
"""byteps_amd flagship benchmark — images/sec for ResNet-50 (default) or
BERT-large data-parallel training on MI355X, synthetic data, random-init
weights (metric and vehicle per BASELINE.json / reference
example/pytorch/benchmark_byteps.py:74-130).

Launched by the driver as:
  python bench.py --gpus N --steps K --warmup W          (N=1)
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

Rank 0 prints exactly ONE JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="resnet50",
                   choices=["resnet50", "vgg16", "bert-large", "mlp"])
    p.add_argument("--batch-size", type=int, default=0,
                   help="per-GPU batch (default 64 — the reference's "
                        "published config, README.md:34-38)")
    p.add_argument("--seq-len", type=int, default=128,
                   help="BERT sequence length (128 = phase-1 pretraining, "
                        "the regime of the reference's batch-64 runs)")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--graph", default="auto", choices=["auto", "on", "off"],
                   help="hipGraph-capture the train step")
    p.add_argument("--device", default="cuda" if torch.cuda.is_available()
                   else "cpu")
    p.add_argument("--compression", default="none",
                   choices=["none", "onebit", "topk", "randomk", "dithering", "fp8"])
    p.add_argument("--partition-mb", type=int, default=0,
                   help="override BPS_PARTITION_BYTES (MiB)")
    p.add_argument("--cross-barrier", action="store_true",
                   help="pipelined per-bucket updates (reference "
                        "cross_barrier.py): optimizer applies as each "
                        "bucket's communication lands; the next forward "
                        "blocks per-layer on its params")
    return p.parse_args()


def build(args, device):
    from byteps_amd import models
    if args.model == "resnet50":
        net = models.resnet50()
        batch = args.batch_size or 64
        x = torch.randn(batch, 3, 224, 224, device=device)
        y = torch.randint(0, 1000, (batch,), device=device)
        if device.type == "cuda":
            net = net.to(memory_format=torch.channels_last)
            x = x.to(memory_format=torch.channels_last)
        loss_fn = torch.nn.CrossEntropyLoss()

        def step_fn(m):
            return loss_fn(m(x), y)
        per_step_items = batch
        unit = "images/sec"
    elif args.model == "vgg16":
        net = models.vgg16()
        batch = args.batch_size or 64
        x = torch.randn(batch, 3, 224, 224, device=device)
        y = torch.randint(0, 1000, (batch,), device=device)
        if device.type == "cuda":
            net = net.to(memory_format=torch.channels_last)
            x = x.to(memory_format=torch.channels_last)
        loss_fn = torch.nn.CrossEntropyLoss()

        def step_fn(m):
            return loss_fn(m(x), y)
        per_step_items = batch
        unit = "images/sec"
    elif args.model == "bert-large":
        net = models.bert_large()
        batch = args.batch_size or 64
        S = args.seq_len
        ids = torch.randint(0, 30522, (batch, S), device=device)
        labels = torch.randint(0, 30522, (batch, S), device=device)

        def step_fn(m):
            mod = m.module if hasattr(m, "module") else m
            return mod.loss(ids, labels)
        per_step_items = batch * S
        unit = "tokens/sec"
    else:  # mlp
        net = models.mnist_mlp()
        batch = args.batch_size or 64
        x = torch.randn(batch, 1, 28, 28, device=device)
        y = torch.randint(0, 10, (batch,), device=device)
        loss_fn = torch.nn.CrossEntropyLoss()

        def step_fn(m):
            return loss_fn(m(x), y)
        per_step_items = batch
        unit = "images/sec"
    net = net.to(device)
    return net, step_fn, per_step_items, unit, batch


def main():
    args = parse_args()
    if args.partition_mb:
        os.environ["BPS_PARTITION_BYTES"] = str(args.partition_mb * 2**20)

    server = None
    if args.compression != "none":
        # BASELINE configs 4/5: compressed gradients through a colocated
        # CPU PS server (port derived from MASTER_PORT so all ranks agree)
        base_port = int(os.environ.get("MASTER_PORT", "29500"))
        ps_port = base_port + 137
        os.environ.setdefault("BPS_FORCE_DISTRIBUTED", "1")
        os.environ.setdefault("BPS_NUM_SERVER", "1")
        os.environ.setdefault("BPS_SERVER_URIS", "127.0.0.1:%d" % ps_port)
        os.environ.setdefault("BPS_MIN_COMPRESS_BYTES", "65536")
        # colocated-PS sweet spot (same-box sweep, profiles/
        # MEASUREMENTS.md): 32 MiB buckets halve the per-bucket server
        # round-trip count; 8/16/64/104 MiB all measured slower
        if not args.partition_mb:
            os.environ.setdefault("BPS_PARTITION_BYTES",
                                  str(32 * 2**20))
        if int(os.environ.get("LOCAL_RANK", "0")) == 0:
            from byteps_amd.ops import core
            server = core().Server(
                ps_port, 8,
                os.environ.get("BPS_SERVER_ENABLE_SCHEDULE", "0") == "1")
            server.start()

    import byteps_amd.torch as bps
    bps.init()
    world = bps.size()
    rank = bps.rank()
    from byteps_amd import common as _C
    device = _C.device() if args.device == "cuda" else torch.device("cpu")
    on_gpu = device.type == "cuda"
    if on_gpu:
        torch.backends.cudnn.benchmark = True

    net, step_fn, per_step_items, unit, batch = build(args, device)

    cparams = None
    if args.compression != "none":
        cparams = {"compressor_type": args.compression,
                   "ef_type": "vanilla"}
        if args.compression in ("topk", "randomk"):
            cparams["compressor_k"] = 4096
        elif args.compression == "dithering":
            # natural s-level partitions (s ≤ 127; small s → sparse
            # Elias wire)
            cparams["compressor_k"] = 8
            cparams["partition"] = "natural"
    cb = None
    if args.cross_barrier:
        from byteps_amd.torch.cross_barrier import CrossBarrier
        opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9,
                              weight_decay=1e-4)
        cb = CrossBarrier(net, opt)
        model = net
    else:
        from byteps_amd.torch.parallel import DistributedDataParallel as DDP
        model = DDP(net, broadcast_buffers=False,
                    compression_params=cparams)
        opt = torch.optim.SGD(net.parameters(), lr=0.1, momentum=0.9,
                              weight_decay=1e-4)

    # pre-tuned hipBLASLt algorithm selections for gfx950 (TunableOp,
    # generated offline by scripts/gen_tunableop.py — tuning is too slow
    # to run inside a timed benchmark)
    tuning_file = os.path.join(
        os.path.dirname(os.path.abspath(__file__)),
        "byteps_amd", "tuning", "tunableop_gfx950.csv")
    if on_gpu and os.path.exists(tuning_file):
        try:
            torch.cuda.tunable.enable(True)
            torch.cuda.tunable.tuning_enable(False)
            torch.cuda.tunable.set_filename(tuning_file)
            torch.cuda.tunable.read_file()
        except Exception as e:
            print("tunableop load failed: %s" % e, file=sys.stderr)

    use_bf16 = args.dtype == "bf16" and on_gpu
    autocast = torch.autocast("cuda", dtype=torch.bfloat16) if use_bf16 \
        else torch.autocast("cpu", enabled=False)

    def train_step():
        if cb is not None:
            cb.zero_grad()
            with autocast:
                loss = step_fn(model)
            loss.backward()
            cb.step()             # returns immediately; updates pipeline
            return loss
        model.zero_grad_buckets()
        with autocast:
            loss = step_fn(model)
        loss.backward()           # engine self-synchronizes on last grad
        opt.step()
        return loss

    # -- hipGraph capture ---------------------------------------------------
    graphed = None
    # auto: capture at N=1 (well-tested, removes launch gaps); stay eager
    # at N>1 — collectives already overlap via async works and the capture
    # gain measured ~0.3% against real replay risk on an untested topology
    want_graph = args.graph == "on" or (
        args.graph == "auto" and on_gpu and world == 1)
    if args.compression != "none" or args.cross_barrier:
        want_graph = False      # host-side KV / poller work per step
    if want_graph and on_gpu:
        try:
            for _ in range(3):      # warm up allocator + RCCL before capture
                train_step()
            torch.cuda.synchronize()
            if world > 1:
                dist.barrier()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                train_step()
            graphed = g
        except Exception as e:
            if rank == 0:
                print("graph capture failed (%s); falling back to eager"
                      % type(e).__name__, file=sys.stderr)
            graphed = None
        if world > 1:
            # replaying a captured collective sequence on only SOME ranks
            # deadlocks — use the graph only if every rank captured
            ok = torch.tensor([1.0 if graphed is not None else 0.0],
                              device=device)
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if ok.item() < 0.5:
                graphed = None

    def run_step():
        if graphed is not None:
            graphed.replay()
        else:
            train_step()

    for _ in range(max(args.warmup, 1)):
        run_step()

    if world > 1:
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    if cb is not None:
        cb.synchronize()          # drain in-flight per-bucket updates
    if on_gpu:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    if world > 1:
        dist.barrier()

    elapsed = t1 - t0
    # MAX over ranks → aggregate uses the slowest rank's clock
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if on_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    value = per_step_items * world * args.steps / elapsed

    if rank == 0:
        out = {
            "metric": unit.replace("/sec", "/sec (whole node)"),
            "value": round(value, 2),
            "unit": unit,
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": batch * world,
                "seq_len": args.seq_len if args.model == "bert-large" else 224,
                "parallelism": "dp%d" % world,
                "graph": graphed is not None,
                "compression": args.compression,
                "cross_barrier": bool(args.cross_barrier),
            },
        }
        print(json.dumps(out))
    if cb is not None:
        cb.stop()
    bps.shutdown()
    if server is not None:
        server.stop()


if __name__ == "__main__":
    main()
