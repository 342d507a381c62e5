"""CPU unit tests for naming, partitioning, config (reference test model:
SURVEY §4)."""

import os

import pytest
import torch

import byteps_amd.common.config as cfg_mod
from byteps_amd.common.naming import (NameRegistry, ServerAssigner, djb2,
                                      partition_key, declared_key_of, part_of)
from byteps_amd.common.partition import plan_partitions, shard_range


def test_name_registry_same_order_same_keys():
    a, b = NameRegistry(), NameRegistry()
    for n in ["w1", "w2", "b1"]:
        a.declare(n)
        b.declare(n)
    for n in ["w1", "w2", "b1"]:
        assert a.key(n) == b.key(n)
    # re-declaring is idempotent
    assert a.declare("w1") == a.key("w1")


def test_name_registry_redeclare_stable():
    r = NameRegistry()
    keys = {n: r.declare(n) for n in ["g1", "g0", "g2"]}
    r.redeclare_all()
    for n, k in keys.items():
        assert r.key(n) == k


def test_partition_keys():
    pk = partition_key(7, 3)
    assert declared_key_of(pk) == 7
    assert part_of(pk) == 3


def test_server_assigner_balances():
    a = ServerAssigner(4)
    for k in range(100):
        a.assign(k, 1000)
    assert max(a.load) - min(a.load) <= 2000
    # deterministic
    b = ServerAssigner(4)
    for k in range(100):
        assert b.assign(k, 1000) == a.table[k]


def test_djb2_known():
    assert djb2(0) == ((5381 * 33) + ord("0")) & 0xFFFFFFFFFFFFFFFF


def test_plan_partitions_packs_and_splits():
    sizes = [100, 200, 5000, 64]
    plans = plan_partitions(sizes, partition_elems=1024, align=64)
    # every element accounted for
    per_param = {}
    for p in plans:
        for s in p.spans:
            per_param[s.param_index] = per_param.get(s.param_index, 0) + s.numel
        assert p.numel <= 1024 + 64
        assert p.numel % 64 == 0
        for s in p.spans:
            assert s.offset % 64 == 0 or s.offset == 0
    assert per_param == {i: sz for i, sz in enumerate(sizes)}
    # priorities strictly decreasing with index
    pris = [p.priority for p in plans]
    assert pris == sorted(pris, reverse=True)
    # large param split across >1 partitions
    p2_parts = [p for p in plans if any(s.param_index == 2 for s in p.spans)]
    assert len(p2_parts) > 1


def test_shard_range():
    assert shard_range(128, 4, 0) == (0, 32)
    assert shard_range(128, 4, 3) == (96, 128)


def test_config_aliases(monkeypatch):
    monkeypatch.setenv("BYTEPS_PARTITION_BYTES", str(4096000))
    monkeypatch.setenv("DMLC_NUM_WORKER", "3")
    c = cfg_mod.Config.from_env()
    assert c.num_workers == 3
    assert c.partition_bytes % cfg_mod.PAGE_SIZE == 0
    assert c.partition_bytes >= 4096000


def test_config_partition_rounding(monkeypatch):
    monkeypatch.setenv("BPS_PARTITION_BYTES", "1000000")
    monkeypatch.setenv("BPS_LOCAL_SIZE", "8")
    c = cfg_mod.Config.from_env()
    assert c.partition_bytes % (8 * cfg_mod.PAGE_SIZE) == 0


def test_mixed_mode_weighted_assignment(monkeypatch):
    monkeypatch.setenv("BPS_ENABLE_MIXED_MODE", "1")
    monkeypatch.setenv("BPS_COLOCATED_SERVERS", "0")
    monkeypatch.setenv("BPS_MIXED_RATIO", "0.5")
    a = ServerAssigner(2)
    for k in range(400):
        a.assign(k, 1000)
    # colocated server 0 gets ~half the standalone server's bytes
    ratio = a.load[0] / a.load[1]
    assert 0.4 < ratio < 0.65, ratio


def test_ops_cpu_wrappers():
    import byteps_amd.ops as K
    t = torch.arange(16, dtype=torch.float32)
    K.scale_(t, 0.5)
    assert torch.allclose(t, torch.arange(16, dtype=torch.float32) * 0.5)
    y = torch.ones(16)
    K.axpy_(y, t, 2.0)
    assert torch.allclose(y, 1 + 2 * t)
    assert abs(K.norm(t, "l1").item() - t.abs().sum().item()) < 1e-4
    assert abs(K.norm(t, "l2").item() - t.norm().item()) < 1e-4
    assert abs(K.norm(t, "max").item() - t.abs().max().item()) < 1e-6


def test_telemetry_window_decay():
    from byteps_amd.common.telemetry import SpeedMeter
    m = SpeedMeter()
    m.record(1_000_000)
    _, mbps = m.speed()
    assert mbps >= 0.0


def test_cpu_topk_select_parallel():
    """Parallel per-thread-heap top-k matches torch.topk's selection set
    (server reply path, replaces the serial partial_sort)."""
    import torch
    from byteps_amd.ops import core
    c = core()
    for n, k in [(1000, 10), (1 << 16, 256), (5000, 5000), (64, 128)]:
        x = torch.randn(n)
        kk = min(k, n)
        idx = torch.empty(kk, dtype=torch.int32)
        val = torch.empty(kk)
        c.cpu_topk_select(x.data_ptr(), n, k, idx.data_ptr(), val.data_ptr())
        _rv, ref_i = torch.topk(x.abs(), kk)
        assert set(idx.tolist()) == set(ref_i.tolist())
        assert torch.allclose(val, x[idx.long()])
        a = x[idx.long()].abs()
        assert bool((a[:-1] >= a[1:]).all())
