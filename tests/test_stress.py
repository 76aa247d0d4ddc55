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
