"""Scheduler rendezvous: server registration, worker discovery, barrier."""

import threading

from byteps_amd.launcher.scheduler import (Scheduler, barrier,
                                           discover_servers,
                                           register_server)


def test_rendezvous_and_barrier():
    sched = Scheduler(port=0, num_servers=2).start()
    try:
        register_server("10.0.0.1", 9100, "127.0.0.1", sched.port)

        found = {}

        def worker():
            found["servers"] = discover_servers("127.0.0.1", sched.port)

        t = threading.Thread(target=worker)
        t.start()
        # worker blocks until the second server registers
        register_server("10.0.0.2", 9100, "127.0.0.1", sched.port)
        t.join(30)
        assert not t.is_alive()
        assert sorted(found["servers"]) == ["10.0.0.1:9100", "10.0.0.2:9100"]

        # duplicate registration is idempotent
        register_server("10.0.0.1", 9100, "127.0.0.1", sched.port)
        assert len(discover_servers("127.0.0.1", sched.port)) == 2

        # 3-party barrier
        done = []

        def bar():
            barrier("127.0.0.1", sched.port, 3)
            done.append(1)

        ts = [threading.Thread(target=bar) for _ in range(3)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(30)
        assert len(done) == 3
    finally:
        sched.stop()
