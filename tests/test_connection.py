"""Cluster bring-up / teardown and SimpleApp RPC (CPU).

Mirrors the upstream dmlc test shapes (test_connection, test_simple_app)
referenced by SURVEY.md §4.
"""
import threading

import numpy as np

import ps_lite_amd as ps
from ps_lite_amd.parallel import launch_local

_PORT = [23000]


def _boot(num_workers=1, num_servers=1):
    _PORT[0] += 7
    ps.setup_env(num_workers, num_servers, root_port=_PORT[0])
    ths = [
        threading.Thread(target=ps.start, kwargs=dict(role=r, device=-1))
        for r in ("scheduler", "server", "worker")
    ]
    [t.start() for t in ths]
    [t.join() for t in ths]


def _down():
    ths = [
        threading.Thread(target=ps.finalize, kwargs=dict(role=r))
        for r in ("scheduler", "server", "worker")
    ]
    [t.start() for t in ths]
    [t.join() for t in ths]
    ps.clear_registry()


def test_connection_and_ids():
    _boot()
    try:
        assert ps.num_workers() == 1
        assert ps.num_servers() == 1
        assert ps.node_id("scheduler") == 1
        assert ps.node_id("server") == 8
        assert ps.node_id("worker") == 9
        assert ps.my_rank("worker") == 0
    finally:
        _down()


def test_barrier_counts():
    _boot()
    try:
        for _ in range(3):
            ths = [
                threading.Thread(target=ps.barrier, args=(r, ps.SCHEDULER_GROUP
                                                          | ps.SERVER_GROUP
                                                          | ps.WORKER_GROUP))
                for r in ("scheduler", "server", "worker")
            ]
            [t.start() for t in ths]
            [t.join() for t in ths]
    finally:
        _down()


def test_simple_app_rpc():
    _boot()
    try:
        server_app = ps.SimpleApp("server", 10, 0)

        def handle(head, body):
            return (body.decode() + f"|head={head}").encode()

        server_app.set_request_handle(handle)
        worker_app = ps.SimpleApp("worker", 10, 0)
        got = []
        worker_app.set_response_handle(lambda head, body: got.append((head, bytes(body))))
        ts = worker_app.request(42, "hello", ps.node_id("server"))
        worker_app.wait(ts)
        assert got == [(42, b"hello|head=42")]
    finally:
        _down()


def _worker_noop(ps_mod, rank):
    return ps_mod.my_rank("worker")


def test_multiprocess_ranks():
    results = launch_local(3, 2, _worker_noop, timeout=180)
    assert sorted(results.values()) == [0, 1, 2]


def _worker_counters(ps_mod, rank):
    w = ps_mod.KVWorker(0, 0)
    keys = np.array([1], dtype=np.uint64)
    vals = np.ones(256, dtype=np.float32)
    ts = w.push(keys, vals, np.array([256], dtype=np.int32))
    w.wait(ts)
    return ps_mod.send_bytes("worker")


def test_byte_counters():
    results = launch_local(1, 1, _worker_counters, timeout=180)
    assert results[0] >= 256 * 4
