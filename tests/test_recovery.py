"""Elastic recovery: a restarted worker inherits a dead worker's id
(ps-lite van.cc:266-320 parity) and the cluster keeps serving."""
import multiprocessing as mp
import os
import random
import sys
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

ENV_BASE = {
    "DMLC_NUM_WORKER": "1",
    "DMLC_NUM_SERVER": "1",
    "DMLC_PS_ROOT_URI": "127.0.0.1",
    "PS_HEARTBEAT_INTERVAL": "1",
    "PS_HEARTBEAT_TIMEOUT": "2",
}


def _run_role(role, port, q, behavior):
    os.environ.update(ENV_BASE)
    os.environ["DMLC_PS_ROOT_PORT"] = str(port)
    sys.path.insert(0, str(REPO))
    import ps_lite_amd as ps

    ps.start(role=role, device=-1)
    if role == "server":
        server = ps.KVServer(0)
        server.set_default_handle()
        q.put(("server", "up"))
        ps.finalize(role=role)  # blocks until everyone else finalizes
        del server
        q.put(("server", "done"))
    elif role == "scheduler":
        q.put(("scheduler", "up"))
        ps.finalize(role=role)
        q.put(("scheduler", "done"))
    elif behavior == "die":
        q.put(("worker", "dying"))
        time.sleep(1)  # let the queue feeder thread flush
        os._exit(0)  # crash without finalize -> becomes a dead node
    else:  # replacement worker
        assert ps.my_rank("worker") == 0, "should inherit the dead worker's rank"
        worker = ps.KVWorker(0, 0)
        keys = np.array([3], dtype=np.uint64)
        vals = np.ones(64, dtype=np.float32)
        ts = worker.push(keys, vals, np.array([64], dtype=np.int32))
        worker.wait(ts)
        out = worker.pull(keys)
        ok = bool(np.allclose(out, vals))
        ps.finalize(role="worker")
        q.put(("worker2", "ok" if ok else "bad"))


def test_worker_recovery():
    port = random.randint(21000, 50000)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = []
    for role, behavior in (("scheduler", None), ("server", None), ("worker", "die")):
        p = ctx.Process(target=_run_role, args=(role, port, q, behavior), daemon=True)
        p.start()
        procs.append(p)
    # wait for the doomed worker to register and die
    seen = {}
    deadline = time.time() + 60
    while "worker" not in seen and time.time() < deadline:
        role, st = q.get(timeout=60)
        seen[role] = st
    assert seen.get("worker") == "dying"
    time.sleep(3)  # exceed the heartbeat timeout
    p = ctx.Process(target=_run_role, args=(("worker"), port, q, "recover"), daemon=True)
    p.start()
    procs.append(p)
    while "worker2" not in seen and time.time() < deadline:
        role, st = q.get(timeout=90)
        seen[role] = st
    assert seen.get("worker2") == "ok", seen
    for p in procs:
        p.join(timeout=30)
