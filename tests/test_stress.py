"""Concurrency stress: multithreaded sessions hammering one KVWorker
(the reference's test_benchmark_stress multithreaded-session pattern,
SURVEY.md §4) plus mixed push/pull traffic across 2x2 processes."""
import numpy as np

from ps_lite_amd.parallel import launch_local


def _threaded_worker(ps, rank):
    import threading

    worker = ps.KVWorker(0, 0)
    nthreads, iters, width = 4, 15, 256
    errors = []

    def session(tid):
        try:
            key = np.array([1000 * (rank + 1) + tid], dtype=np.uint64)
            lens = np.array([width], dtype=np.int32)
            acc = np.zeros(width, dtype=np.float32)
            for it in range(iters):
                vals = np.full(width, float(tid + it + 1), dtype=np.float32)
                worker.wait(worker.push(key, vals, lens))
                acc += vals
                out = worker.pull(key)
                if not np.allclose(out, acc):
                    errors.append((tid, it, out[:3].tolist(), acc[:3].tolist()))
                    return
        except Exception as e:  # noqa
            errors.append((tid, repr(e)))

    threads = [threading.Thread(target=session, args=(t,)) for t in range(nthreads)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    return errors


def test_multithreaded_sessions():
    results = launch_local(2, 2, _threaded_worker, timeout=300)
    for rank, errors in results.items():
        assert errors == [], (rank, errors)


def _dense_reduce_worker(ps, rank):
    """Both workers accumulate into the SAME key each step and read back
    the cluster-wide sum (test_benchmark_stress DenseReduce shape)."""
    w = ps.KVWorker(0, 0)
    key = np.array([77], dtype=np.uint64)
    width = 512
    lens = np.array([width], dtype=np.int32)
    expect = np.zeros(width, dtype=np.float32)
    bad = []
    for step in range(8):
        mine = np.full(width, float(rank + 1 + step), dtype=np.float32)
        other = np.full(width, float((1 - rank) + 1 + step), dtype=np.float32)
        w.wait(w.push(key, mine, lens))
        expect += mine + other
        ps.barrier("worker", ps.WORKER_GROUP)  # both pushes landed
        out = w.pull(key)
        if not np.allclose(out, expect):
            bad.append((step, out[:2].tolist(), expect[:2].tolist()))
        ps.barrier("worker", ps.WORKER_GROUP)  # reads done before next round
    return bad


def test_dense_reduce_two_workers():
    results = launch_local(2, 2, _dense_reduce_worker, timeout=300)
    for rank, bad in results.items():
        assert bad == [], (rank, bad)


def _reduce_chaos_worker(ps, rank):
    """4 workers x 4 keys x 8 reduce rounds with random per-worker
    delays: exercises every deferral path of the round protocol (pushes
    arriving before the previous round drained, next-round pulls,
    replay after reset) under chaotic timing."""
    import random
    import time as _t

    nw, nkeys, rounds, width = 4, 4, 8, 64
    server = ps.KVServer(0)
    server.set_reduce_handle(num_workers=nw)
    ps.barrier("worker", ps.WORKER_GROUP)
    w = ps.KVWorker(0, 0)
    rng = random.Random(1000 + rank)
    step = (1 << 64) // nw
    keys = [np.array([s * step + 7], dtype=np.uint64) for s in range(nkeys)]
    lens = np.array([width], dtype=np.int32)
    bad = []
    for r in range(rounds):
        tss = []
        outs = []
        for k in range(nkeys):
            if rng.random() < 0.5:
                _t.sleep(rng.random() * 0.01)
            val = float((rank + 1) * 1000 + r)
            tss.append(w.push(keys[k], np.full(width, val, dtype=np.float32), lens))
            outs.append(w.pull(keys[k]))  # overlapped; held until all pushed
        for ts in tss:
            w.wait(ts)
        expect = sum((i + 1) * 1000 + r for i in range(nw))
        for k, out in enumerate(outs):
            if not np.allclose(out, float(expect)):
                bad.append((r, k, float(out[0]), expect))
    return bad, server


def test_reduce_round_chaos_four_workers():
    results = launch_local(4, 4, _reduce_chaos_worker, joint=True, timeout=420)
    for rank, bad in results.items():
        assert bad == [], (rank, bad)
