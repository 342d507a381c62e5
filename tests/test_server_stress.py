"""KV server robustness: many keys, concurrent clients, interleaved
rounds, barrier service, reconnect behavior."""

import struct
import threading

import numpy as np
import pytest
import torch


@pytest.fixture()
def core():
    from byteps_amd.ops import _core
    return _core


def _init_key(core, kv, key, n, expected, codec=0, server=0):
    payload = np.zeros(24, dtype=np.uint8)
    struct.pack_into("<QIIII", payload, 0, n, expected, 0, 0, 0)
    t = kv.submit(server, core.OP_INIT, key, payload.ctypes.data, 24, 0, 0,
                  core.make_cmd(codec, 0, False), 0)
    kv.wait(t)


def test_many_keys_many_rounds(core):
    srv = core.Server(0, 4, False)
    srv.start()
    try:
        kv = core.KVClient(0, ["127.0.0.1:%d" % srv.port])
        n = 1024
        nkeys = 32
        rounds = 5
        bufs = {k: (np.random.randn(n).astype(np.float32),
                    np.zeros(n, dtype=np.float32)) for k in range(nkeys)}
        for k in bufs:
            _init_key(core, kv, k, n, 1)
        for r in range(1, rounds + 1):
            tickets = []
            for k, (send, _recv) in bufs.items():
                tickets.append(kv.submit(0, core.OP_PUSH, k,
                                         send.ctypes.data, n * 4, 0, 0, 0, r))
            for t in tickets:
                kv.wait(t)
            tickets = []
            for k, (_send, recv) in bufs.items():
                tickets.append(kv.submit(0, core.OP_PULL, k, 0, 0,
                                         recv.ctypes.data, n * 4, 0, r))
            for t in tickets:
                ln, ver = kv.wait(t)
                assert ln == n * 4 and ver == r
            for k, (send, recv) in bufs.items():
                np.testing.assert_allclose(recv, send, rtol=1e-6)
        kv.close()
    finally:
        srv.stop()


def test_concurrent_clients_sum(core):
    srv = core.Server(0, 4, True)     # scheduling enabled
    srv.start()
    try:
        nclients, n = 4, 4096
        datas = [np.full(n, float(i + 1), dtype=np.float32)
                 for i in range(nclients)]
        recvs = [np.zeros(n, dtype=np.float32) for _ in range(nclients)]
        errs = []

        def worker(i):
            try:
                kv = core.KVClient(i, ["127.0.0.1:%d" % srv.port])
                _init_key(core, kv, 99, n, nclients)
                t = kv.submit(0, core.OP_PUSH, 99, datas[i].ctypes.data,
                              n * 4, 0, 0, 0, 1)
                kv.wait(t)
                t = kv.submit(0, core.OP_PULL, 99, 0, 0,
                              recvs[i].ctypes.data, n * 4, 0, 1)
                ln, ver = kv.wait(t)
                assert ln == n * 4
                kv.close()
            except Exception as e:      # pragma: no cover
                errs.append(e)

        threads = [threading.Thread(target=worker, args=(i,))
                   for i in range(nclients)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(60)
        assert not errs, errs
        expect = sum(range(1, nclients + 1))
        for r in recvs:
            np.testing.assert_allclose(r, expect, rtol=1e-6)
    finally:
        srv.stop()


def test_pull_before_push_defers(core):
    srv = core.Server(0, 2, False)
    srv.start()
    try:
        kv = core.KVClient(0, ["127.0.0.1:%d" % srv.port])
        n = 256
        _init_key(core, kv, 7, n, 1)
        recv = np.zeros(n, dtype=np.float32)
        # pull for round 1 before any push — must block until merge
        t_pull = kv.submit(0, core.OP_PULL, 7, 0, 0, recv.ctypes.data,
                           n * 4, 0, 1)
        assert not kv.test(t_pull)
        send = np.arange(n, dtype=np.float32)
        t_push = kv.submit(0, core.OP_PUSH, 7, send.ctypes.data, n * 4,
                           0, 0, 0, 1)
        kv.wait(t_push)
        ln, ver = kv.wait(t_pull)
        assert ver == 1
        np.testing.assert_allclose(recv, send)
        kv.close()
    finally:
        srv.stop()


def test_unknown_key_errors(core):
    srv = core.Server(0, 2, False)
    srv.start()
    try:
        kv = core.KVClient(0, ["127.0.0.1:%d" % srv.port])
        buf = np.zeros(16, dtype=np.float32)
        t = kv.submit(0, core.OP_PULL, 12345, 0, 0, buf.ctypes.data, 64,
                      0, 1)
        ln, aux = kv.wait(t)
        assert aux == (1 << 64) - 1     # error marker
        kv.close()
    finally:
        srv.stop()
