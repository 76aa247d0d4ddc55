"""KV push/pull correctness over the localhost TCP van (CPU).

Restores the upstream-style unit coverage the byteps branch dropped
(test_connection / test_kv_app / test_kv_app_multi_workers).
"""
import threading

import numpy as np
import pytest

import ps_lite_amd as ps
from ps_lite_amd.parallel import launch_local

_PORT = [21000]


def _inproc_cluster(num_workers=1, num_servers=1):
    _PORT[0] += 7
    ps.setup_env(num_workers, num_servers, root_port=_PORT[0])
    ths = [
        threading.Thread(target=ps.start, kwargs=dict(role=r, device=-1))
        for r in ("scheduler", "server", "worker")
    ]
    [t.start() for t in ths]
    [t.join() for t in ths]


def _inproc_teardown():
    ths = [
        threading.Thread(target=ps.finalize, kwargs=dict(role=r))
        for r in ("scheduler", "server", "worker")
    ]
    [t.start() for t in ths]
    [t.join() for t in ths]
    ps.clear_registry()


def test_push_pull_default_handle():
    _inproc_cluster()
    try:
        server = ps.KVServer(0)
        server.set_default_handle()
        worker = ps.KVWorker(0, 0)
        keys = np.array([1, 5, 9], dtype=np.uint64)
        vals = np.arange(9, dtype=np.float32)
        lens = np.array([3, 3, 3], dtype=np.int32)
        n_push = 4
        tss = [worker.push(keys, vals, lens) for _ in range(n_push)]
        for ts in tss:
            worker.wait(ts)
        out = worker.pull(keys)
        assert np.allclose(out, n_push * vals)
    finally:
        _inproc_teardown()


def test_python_handle_assign():
    _inproc_cluster()
    try:
        store = {}

        def handle(meta, keys, vals):
            if meta["push"]:
                off = 0
                k = len(vals) // len(keys)
                for key in keys:
                    store[int(key)] = vals[off:off + k].copy()
                    off += k
                return None
            return np.concatenate([store[int(key)] for key in keys])

        server = ps.KVServer(0)
        server.set_python_handle(handle)
        worker = ps.KVWorker(0, 0)
        keys = np.array([2, 4], dtype=np.uint64)
        vals = np.array([1.0, 2.0, 3.0, 4.0], dtype=np.float32)
        ts = worker.push(keys, vals, np.array([2, 2], dtype=np.int32))
        worker.wait(ts)
        out = worker.pull(keys)
        assert np.allclose(out, vals)
    finally:
        _inproc_teardown()


def test_zpush_zpull_host_buffers():
    _inproc_cluster()
    try:
        server = ps.KVServer(0)
        server.set_default_handle()
        worker = ps.KVWorker(0, 0)
        keys = np.array([7], dtype=np.uint64)
        vals = np.full(1024, 2.5, dtype=np.float32)
        ts = worker.zpush_ptr(keys, vals.ctypes.data, vals.nbytes, -1,
                              np.array([1024], dtype=np.int32))
        worker.wait(ts)
        dst = np.zeros(1024, dtype=np.float32)
        ts = worker.zpull_ptr(keys, dst.ctypes.data, dst.nbytes, -1,
                              np.array([1024], dtype=np.int32))
        worker.wait(ts)
        assert np.allclose(dst, vals)
    finally:
        _inproc_teardown()


# ---------------- multi-process tests (the reference's local.sh pattern) ---


def _worker_push_pull(ps_mod, rank):
    worker = ps_mod.KVWorker(0, 0)
    num_servers = ps_mod.num_servers()
    # keys spanning every server's range: server i owns [i*2^64/S, ...)
    step = (1 << 64) // num_servers
    keys = np.array([i * step + 17 for i in range(num_servers)], dtype=np.uint64)
    vals = (np.arange(8 * num_servers) + rank).astype(np.float32)
    lens = np.full(num_servers, 8, dtype=np.int32)
    ts = worker.push(keys, vals, lens)
    worker.wait(ts)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    out = worker.pull(keys)
    return out.tolist()


def test_multiprocess_2workers_2servers():
    results = launch_local(2, 2, _worker_push_pull, timeout=180)
    assert set(results.keys()) == {0, 1}
    # sum over both workers: vals_r = arange(16) + r => total = 2*arange(16)+1
    expect = 2 * np.arange(16, dtype=np.float32) + 1
    for rank, out in results.items():
        assert np.allclose(np.array(out), expect), (rank, out)


def _worker_single(ps_mod, rank):
    worker = ps_mod.KVWorker(0, 0)
    keys = np.array([3], dtype=np.uint64)
    vals = np.ones(64, dtype=np.float32) * (rank + 1)
    ts = worker.push(keys, vals, np.array([64], dtype=np.int32))
    worker.wait(ts)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    return worker.pull(keys).tolist()


def _joint_fn(ps_mod, rank):
    server = ps_mod.KVServer(0)
    server.set_default_handle()
    # one worker instance per joint process: a worker-group barrier
    # synchronizes every process after server creation
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    out = _worker_single(ps_mod, rank)
    return out, server


def test_multiprocess_joint_mode():
    results = launch_local(2, 2, _joint_fn, joint=True, timeout=180)
    expect = np.full(64, 3.0)  # ranks 1+2
    for rank, out in results.items():
        assert np.allclose(np.array(out), expect)


def _host_pool_worker(ps_mod, rank):
    # large host payloads from the host-shm pool: by-ref push + in-place
    # pull across BOTH servers (exercises per-slice addr offsets)
    worker = ps_mod.KVWorker(0, 0)
    n = 1 << 16  # 256 KiB per key, well above the inline threshold
    step = (1 << 64) // 2
    keys = np.array([9, step + 9], dtype=np.uint64)
    lens = np.array([n, n], dtype=np.int32)
    src = ps_mod.host_alloc(2 * n * 4)
    dst = ps_mod.host_alloc(2 * n * 4)
    vals = np.concatenate([np.full(n, 3.0, dtype=np.float32),
                           np.full(n, 5.0, dtype=np.float32)])
    src.copy_from(vals)
    worker.wait(worker.zpush_ptr(keys, src.ptr, 2 * n * 4, -1, lens, cmd=1))
    worker.wait(worker.zpull_ptr(keys, dst.ptr, 2 * n * 4, -1, lens, cmd=1))
    out = dst.to_numpy_f32()
    return [float(out[0]), float(out[n - 1]), float(out[n]), float(out[-1])]


def test_host_pool_zero_copy_two_servers():
    results = launch_local(1, 2, _host_pool_worker, timeout=180)
    assert results[0] == [3.0, 3.0, 5.0, 5.0], results


def test_resend_with_drop():
    # force pure TCP: drop injection + retransmission are TCP-path
    # features (the shm plane is lossless and would bypass both)
    env = {"PS_RESEND": "1", "PS_RESEND_TIMEOUT": "200", "PS_DROP_MSG": "10",
           "XPS_HOST_PLANE": "0"}
    results = launch_local(1, 1, _worker_single, env_extra=env, timeout=180)
    assert np.allclose(np.array(results[0]), np.ones(64))


def _random_shard_worker(ps_mod, rank):
    """Random keys + non-uniform lens across 4 servers: exercises the
    slicer boundaries and the worker-side pull merge (re-sorting slices,
    lens bookkeeping) against a local reference."""
    server = None  # plain worker (separate server procs use default fn)
    w = ps_mod.KVWorker(0, 0)
    rng = np.random.default_rng(4242)
    nkeys = 57
    keys = np.sort(rng.choice(1 << 62, size=nkeys, replace=False)).astype(np.uint64)
    lens = rng.integers(1, 96, size=nkeys).astype(np.int32)
    vals = rng.standard_normal(int(lens.sum())).astype(np.float32)
    w.wait(w.push(keys, vals, lens))
    got = w.pull(keys)
    assert got.shape == vals.shape
    assert np.allclose(got, vals, atol=1e-6)
    # partial pull of a random subset must return that subset's slices
    idx = np.sort(rng.choice(nkeys, size=13, replace=False))
    sub_keys = keys[idx]
    expect = np.concatenate([
        vals[int(lens[:i].sum()):int(lens[:i].sum() + lens[i])] for i in idx])
    got2 = w.pull(sub_keys)
    assert np.allclose(got2, expect, atol=1e-6)
    return True


def test_random_sharding_four_servers():
    results = launch_local(1, 4, _random_shard_worker, timeout=240)
    assert results[0] is True


def _zero_copy_worker(ps_mod, rank):
    """Transport-level zero-copy assertion (reference
    test_benchmark.cc:169-181 pointer-equality check): pushes from the
    host pool must arrive at the server BY REFERENCE, not staged."""
    w = ps_mod.KVWorker(0, 0)
    before = ps_mod._core.zero_copy_recv_count()
    n = 1 << 16  # 256 KB > inline budget -> must ride by-ref
    buf = ps_mod.host_alloc(n * 4)
    buf.copy_from(np.ones(n, dtype=np.float32))
    keys = np.array([11], dtype=np.uint64)
    lens = np.array([n], dtype=np.int32)
    w.wait(w.zpush_ptr(keys, buf.ptr, n * 4, -1, lens, cmd=2))
    out = w.pull(keys)
    assert np.allclose(out, 1.0)
    # the worker-side counter counts its own received by-ref responses;
    # the PUSH was received by the server process, so probe via a second
    # push from this process to itself? No: in separate-process mode we
    # can only check our own receptions — the pull response must have
    # been delivered in place (kOptInPlace) or by-ref
    return int(ps_mod._core.zero_copy_recv_count() - before)


def test_zero_copy_reception():
    results = launch_local(1, 1, _zero_copy_worker, timeout=180)
    # worker side alone sees >= 0; the real assertion runs in joint mode
    # below where worker and server share a process
    assert results[0] >= 0


def _zero_copy_joint_worker(ps_mod, rank):
    server = ps_mod.KVServer(0)
    server.set_default_handle()
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    w = ps_mod.KVWorker(0, 0)
    before = ps_mod._core.zero_copy_recv_count()
    n = 1 << 16
    buf = ps_mod.host_alloc(n * 4)
    buf.copy_from(np.full(n, 2.0, dtype=np.float32))
    keys = np.array([12], dtype=np.uint64)
    lens = np.array([n], dtype=np.int32)
    w.wait(w.zpush_ptr(keys, buf.ptr, n * 4, -1, lens, cmd=2))
    got = w.pull(keys)
    assert np.allclose(got, 2.0)
    after = ps_mod._core.zero_copy_recv_count()
    # joint process: the server's reception of our 256 KB push MUST have
    # been by-reference into the mapped host pool (zero-copy)
    assert after - before >= 1, (before, after)
    return True, server


def test_zero_copy_reception_joint():
    results = launch_local(1, 1, _zero_copy_joint_worker, joint=True, timeout=180)
    assert results[0] is True


def _cpu_reduce_worker(ps_mod, rank):
    """BytePS reduce rounds on CPU (KVServerReduceHandle): overlapped
    push+pull, no inter-worker barrier — same protocol as the GPU
    reduce handler, testable without a GPU."""
    server = ps_mod.KVServer(0)
    server.set_reduce_handle(num_workers=2)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    w = ps_mod.KVWorker(0, 0)
    n = 2048
    keys = np.array([21], dtype=np.uint64)
    lens = np.array([n], dtype=np.int32)
    outs = []
    for step in range(5):  # rounds exercise reset + deferred replay
        vals = np.full(n, float(rank + 1 + step), dtype=np.float32)
        ts1 = w.push(keys, vals, lens)
        out = w.pull(keys)  # overlapped: pull issued while pushes in flight
        w.wait(ts1)
        outs.append(float(out[0]))
        assert np.allclose(out, out[0])
    return outs, server


def test_cpu_reduce_rounds_two_workers():
    results = launch_local(2, 2, _cpu_reduce_worker, joint=True, timeout=300)
    # round k: workers push (1+k) and (2+k) -> both pull 3+2k
    for rank, outs in results.items():
        assert outs == [3.0 + 2 * k for k in range(5)], results


def _cpu_reduce_multikey_worker(ps_mod, rank):
    """Bucketed reduce rounds: ONE multi-key message per round (the
    BytePS DenseReduce batching — group round accounting, not per-key).
    Mixed bucket lengths exercise the per-key offsets."""
    server = ps_mod.KVServer(0)
    server.set_reduce_handle(num_workers=2)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    w = ps_mod.KVWorker(0, 0)
    lens = np.array([64, 256, 128], dtype=np.int32)
    keys = np.array([31, 32, 33], dtype=np.uint64)
    total = int(lens.sum())
    outs = []
    for step in range(4):
        vals = np.full(total, float(rank + 1 + step), dtype=np.float32)
        ts1 = w.push(keys, vals, lens)
        out = w.pull(keys)  # overlapped: pull issued while pushes in flight
        w.wait(ts1)
        assert out.shape[0] == total
        assert np.allclose(out, out[0]), out[:4].tolist()
        outs.append(float(out[0]))
    return outs, server


def test_cpu_reduce_rounds_multikey():
    results = launch_local(2, 2, _cpu_reduce_multikey_worker, joint=True, timeout=300)
    for rank, outs in results.items():
        assert outs == [3.0 + 2 * k for k in range(4)], results


def _cpu_reduce_deferred_push_worker(ps_mod, rank):
    """Deferred next-round replay: the worker pushes round k+1 BEFORE
    pulling round k, so when the round's last pull arrives the deferred
    push replays inside the pull path (the code path that double-locked
    the GPU handler's mutex before the round-2 fix)."""
    server = ps_mod.KVServer(0)
    server.set_reduce_handle(num_workers=1)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    w = ps_mod.KVWorker(0, 0)
    keys = np.array([77], dtype=np.uint64)
    n = 512
    lens = np.array([n], dtype=np.int32)
    ts1 = w.push(keys, np.full(n, 5.0, dtype=np.float32), lens)
    # round-2 push sent BEFORE the round-1 pull: per-sender FIFO delivers
    # it first, so it lands in waiting_pushes until the pull drains
    ts2 = w.push(keys, np.full(n, 9.0, dtype=np.float32), lens)
    out1 = w.pull(keys)   # releases round 1, replays the deferred push
    out2 = w.pull(keys)   # round 2 value
    w.wait(ts1)
    w.wait(ts2)
    return [float(out1[0]), float(out2[0])], server


def test_cpu_reduce_deferred_push_replay():
    results = launch_local(1, 1, _cpu_reduce_deferred_push_worker, joint=True, timeout=240)
    assert results[0] == [5.0, 9.0], results


def _fused_pushpull_worker(ps_mod, rank):
    """ZPushPull: one request carries the push AND returns the post-push
    values (halves the sparse round's trips). CPU default handle: the
    response must equal the accumulated store."""
    server = ps_mod.KVServer(0)
    server.set_default_handle()
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    w = ps_mod.KVWorker(0, 0)
    n = 1024
    keys = np.array([61, 62], dtype=np.uint64)
    lens = np.array([n, n], dtype=np.int32)
    push = ps_mod.host_alloc(2 * n * 4)
    out = ps_mod.host_alloc(2 * n * 4)
    vals = np.concatenate([np.full(n, 2.0, dtype=np.float32),
                           np.full(n, 5.0, dtype=np.float32)])
    push.copy_from(vals)
    for it in range(3):  # sum handle: round k returns (k+1)*vals
        w.wait(w.zpushpull_ptr(keys, push.ptr, out.ptr, 2 * n * 4, -1, lens, cmd=2))
        got = out.to_numpy_f32()
        assert np.allclose(got, (it + 1) * vals), (it, got[:3].tolist())
    return True


def test_fused_pushpull_cpu():
    results = launch_local(1, 1, _fused_pushpull_worker, joint=True, timeout=240)
    assert results[0] is True


def _ordering_worker(ps_mod, rank):
    """Cross-transport FIFO: an 8 KB push rides the TCP fallback (host
    heap, > inline budget) while the pull request rides the shm ring —
    without sequence gating the pull overtakes the push and reads
    zeros/stale. Meta.seq + the receive-side gate must serialize them."""
    w = ps_mod.KVWorker(0, 0)
    n = 2048  # 8 KB > kInlineMax
    lens = np.array([n], dtype=np.int32)
    for it in range(10):
        keys = np.array([500 + it], dtype=np.uint64)
        vals = np.full(n, float(it + 1), dtype=np.float32)
        ts = w.push(keys, vals, lens)  # fresh key: sum(0 + vals) == vals
        got = w.pull(keys)  # issued immediately: must NOT overtake the push
        w.wait(ts)
        assert np.allclose(got, vals), (it, got[:3].tolist())
    return True


def test_cross_transport_ordering():
    results = launch_local(1, 1, _ordering_worker, timeout=240)
    assert results[0] is True


def _mixed_drop_worker(ps_mod, rank):
    """Mixed transports + fault injection: small messages ride the
    lossless shm plane, 8 KB ones the TCP fallback where PS_DROP_MSG
    drops some. A dropped TCP push leaves a sequence gap; the gate must
    HOLD later ring messages until the resend fills it — order and
    values both intact."""
    w = ps_mod.KVWorker(0, 0)
    big, small = 2048, 128  # 8 KB (TCP) / 512 B (inline -> ring)
    total = 0.0
    for it in range(12):
        n = big if it % 2 == 0 else small
        keys = np.array([900], dtype=np.uint64)
        lens = np.array([n], dtype=np.int32)
        w.wait(w.push(keys, np.full(n, 1.0, dtype=np.float32), lens))
        total += 1.0
    out = w.pull(np.array([900], dtype=np.uint64))
    # the store entry grows to `big`; elements [0, small) saw every push,
    # the tail only the big ones
    assert out.shape[0] == big
    assert np.allclose(out[:small], total), out[:4].tolist()
    assert np.allclose(out[small:], 6.0), out[small:small + 4].tolist()
    return True


def test_ordering_with_drops_and_resend():
    env = {"PS_RESEND": "1", "PS_RESEND_TIMEOUT": "200", "PS_DROP_MSG": "20"}
    results = launch_local(1, 1, _mixed_drop_worker, env_extra=env, timeout=240)
    assert results[0] is True
