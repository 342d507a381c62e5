"""BERT-large throughput floor on one MI355X — carries a driver-observed
BERT datapoint in GPUTEST.  The reference's
headline metric is BERT-large scaling (README.md:34-38); its absolute
per-GPU rate here is measured by bench.py --model bert-large
(185.6k tokens/s on 2026-09 boxes, profiles/MEASUREMENTS.md) — the floor
asserts we stay within ~35% of that, loose enough for box variance."""

import time

import pytest
import torch

pytestmark = pytest.mark.gpu

FLOOR_TOKENS_PER_S = 120_000
BATCH, SEQ = 64, 128


def test_bert_large_tokens_per_sec_floor():
    from byteps_amd import models
    net = models.bert_large().to("cuda")
    ids = torch.randint(0, 30522, (BATCH, SEQ), device="cuda")
    labels = torch.randint(0, 30522, (BATCH, SEQ), device="cuda")
    opt = torch.optim.SGD(net.parameters(), lr=0.01, momentum=0.9)

    def step():
        opt.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = net.loss(ids, labels)
        loss.backward()
        opt.step()
        return loss

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n_steps = 5
    for _ in range(n_steps):
        step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tps = BATCH * SEQ * n_steps / dt
    assert tps > FLOOR_TOKENS_PER_S, \
        "BERT-large %.0f tokens/s below floor %d (%.1f ms/step)" % (
            tps, FLOOR_TOKENS_PER_S, dt / n_steps * 1e3)
