"""Instance groups (DMLC_GROUP_SIZE) and multi-customer workers —
upstream test_kv_app_multi_workers / §2.2 parity."""
import numpy as np

from ps_lite_amd.parallel import launch_local


def _multi_customer_worker(ps, rank):
    # two customers on one worker instance, interleaved requests
    w0 = ps.KVWorker(0, 0)
    w1 = ps.KVWorker(0, 1)
    keys = np.array([4], dtype=np.uint64)
    lens = np.array([8], dtype=np.int32)
    v0 = np.ones(8, dtype=np.float32)
    v1 = 2 * np.ones(8, dtype=np.float32)
    ts0 = w0.push(keys, v0, lens)
    ts1 = w1.push(keys, v1, lens)
    w0.wait(ts0)
    w1.wait(ts1)
    out0 = w0.pull(keys)
    out1 = w1.pull(keys)
    return [out0.tolist(), out1.tolist()]


def test_multi_customer_worker():
    results = launch_local(1, 1, _multi_customer_worker, timeout=180)
    out0, out1 = results[0]
    assert np.allclose(np.array(out0), 3.0)  # both customers' pushes summed
    assert np.allclose(np.array(out1), 3.0)


def test_group_size_two():
    """One joint process with DMLC_GROUP_SIZE=2 hosts worker instances
    0,1 + server instances 0,1 (4 vans) — ps-lite §2.2 instance groups."""
    import threading

    import ps_lite_amd as ps

    ps.setup_env(2, 2, root_port=28731, DMLC_GROUP_SIZE=2)
    ths = [threading.Thread(target=ps.start, kwargs=dict(role="scheduler", device=-1)),
           threading.Thread(target=ps.start, kwargs=dict(role="joint", device=-1))]
    [t.start() for t in ths]
    [t.join() for t in ths]
    try:
        servers = []
        for g in range(2):
            s = ps.KVServer(0, instance_idx=g)
            s.set_default_handle()
            servers.append(s)
        assert {ps.node_id("server", idx=0), ps.node_id("server", idx=1)} == {8, 10}
        assert {ps.node_id("worker", idx=0), ps.node_id("worker", idx=1)} == {9, 11}
        step = (1 << 64) // 2
        keys = np.array([3, step + 3], dtype=np.uint64)  # one key per server
        lens = np.array([4, 4], dtype=np.int32)
        workers = []
        for g in range(2):
            w = ps.KVWorker(0, 0, instance_idx=g)
            vals = np.full(8, float(g + 1), dtype=np.float32)
            w.wait(w.push(keys, vals, lens))
            workers.append(w)
        out = workers[0].pull(keys)
        assert np.allclose(out, 3.0), out  # instances pushed 1 and 2
    finally:
        ths = [threading.Thread(target=ps.finalize, kwargs=dict(role="scheduler")),
               threading.Thread(target=ps.finalize, kwargs=dict(role="joint"))]
        [t.start() for t in ths]
        [t.join() for t in ths]
        ps.clear_registry()
