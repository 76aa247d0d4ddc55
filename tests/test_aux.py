"""Aux subsystems: checkpoint/resume and message tracing
(ENABLE_PROFILING parity with ps-lite src/van.cc:38-77)."""
import glob
import os

import numpy as np

from ps_lite_amd.parallel import launch_local


def _ckpt_worker_fn(ps, rank):
    worker = ps.KVWorker(0, 0)
    keys = np.array([11, 22], dtype=np.uint64)
    vals = np.arange(8, dtype=np.float32)
    lens = np.array([4, 4], dtype=np.int32)
    worker.wait(worker.push(keys, vals, lens))
    out1 = worker.pull(keys).tolist()
    ps.barrier("worker", ps.SERVER_GROUP | ps.WORKER_GROUP)  # server checkpoints
    ps.barrier("worker", ps.SERVER_GROUP | ps.WORKER_GROUP)  # server reloaded
    out2 = worker.pull(keys).tolist()
    # NB: a 2-tuple return would be treated as (payload, keepalive)
    return [out1, out2]


def _ckpt_server_fn(ps, rank):
    tmpdir = os.environ["XPS_TEST_TMPDIR"]
    path = os.path.join(tmpdir, f"ckpt_{rank}.bin")
    server = ps.KVServer(0)
    server.set_default_handle()
    ps.barrier("server", ps.SERVER_GROUP | ps.WORKER_GROUP)  # worker done pushing
    server.save_checkpoint(path)
    server.load_checkpoint(path)
    ps.barrier("server", ps.SERVER_GROUP | ps.WORKER_GROUP)
    return None, server


def test_checkpoint_roundtrip(tmp_path):
    env = {"XPS_TEST_TMPDIR": str(tmp_path)}
    results = launch_local(1, 1, _ckpt_worker_fn, server_fn=_ckpt_server_fn,
                           env_extra=env, timeout=180)
    out1, out2 = results[0]
    assert np.allclose(np.array(out1), np.arange(8, dtype=np.float32))
    assert out1 == out2  # state identical after save + load
    ckpts = glob.glob(os.path.join(str(tmp_path), "ckpt_*.bin"))
    assert len(ckpts) == 1 and os.path.getsize(ckpts[0]) > 8 * 8


def _trace_worker(ps, rank):
    worker = ps.KVWorker(0, 0)
    keys = np.array([1], dtype=np.uint64)
    vals = np.ones(16, dtype=np.float32)
    worker.wait(worker.push(keys, vals, np.array([16], dtype=np.int32)))
    worker.pull(keys)
    return True


def test_profiling_trace_file(tmp_path):
    prefix = str(tmp_path / "trace")
    env = {"ENABLE_PROFILING": "1", "PROFILE_PATH": prefix}
    results = launch_local(1, 1, _trace_worker, env_extra=env, timeout=180)
    assert results[0] is True
    traces = glob.glob(prefix + "_*")
    assert traces, "no trace files written"
    content = "".join(open(t).read() for t in traces)
    assert "van_send_push" in content or "van_recv_push" in content, content[:500]


def _timing_worker(ps_mod, rank):
    w = ps_mod.KVWorker(0, 0)
    keys = np.array([3], dtype=np.uint64)
    vals = np.ones(512, dtype=np.float32)
    for _ in range(5):
        w.wait(w.push(keys, vals, np.array([512], dtype=np.int32)))
    return True


def test_stage_timing_table(capfd):
    """XPS_TIMING=1 prints the per-stage table at plane shutdown."""
    results = launch_local(1, 1, _timing_worker,
                           env_extra={"XPS_TIMING": "1"}, timeout=180)
    assert results[0] is True
    err = capfd.readouterr().err
    assert "stage timing" in err and "plane_send" in err, err[-2000:]


def _reduce_ckpt_worker(ps, rank):
    worker = ps.KVWorker(0, 0)
    keys = np.array([33], dtype=np.uint64)
    vals = np.full(16, 5.0, dtype=np.float32)
    lens = np.array([16], dtype=np.int32)
    worker.wait(worker.push(keys, vals, lens))
    out1 = worker.pull(keys).tolist()  # completes the 1-worker round
    ps.barrier("worker", ps.SERVER_GROUP | ps.WORKER_GROUP)  # server ckpts
    ps.barrier("worker", ps.SERVER_GROUP | ps.WORKER_GROUP)  # reloaded as default
    out2 = worker.pull(keys).tolist()  # default handle serves it directly
    return [out1, out2]


def _reduce_ckpt_server(ps, rank):
    tmpdir = os.environ["XPS_TEST_TMPDIR"]
    path = os.path.join(tmpdir, "reduce_ckpt.bin")
    server = ps.KVServer(0)
    server.set_reduce_handle(num_workers=1)
    ps.barrier("server", ps.SERVER_GROUP | ps.WORKER_GROUP)
    server.save_checkpoint(path)
    # the on-disk format is shared: swap the SAME server to the default
    # handle and reload (reduce rounds are transient; the stored KV is
    # what counts)
    server.set_default_handle()
    server.load_checkpoint(path)
    ps.barrier("server", ps.SERVER_GROUP | ps.WORKER_GROUP)
    return None, server


def test_reduce_checkpoint_into_default_handle(tmp_path):
    env = {"XPS_TEST_TMPDIR": str(tmp_path)}
    results = launch_local(1, 1, _reduce_ckpt_worker, server_fn=_reduce_ckpt_server,
                           env_extra=env, timeout=180)
    out1, out2 = results[0]
    assert np.allclose(out1, 5.0) and np.allclose(out2, 5.0)
