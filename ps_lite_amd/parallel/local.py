"""Local multi-process cluster launcher.

Reference parity: ps-lite tests/local.sh + tracker/dmlc_local.py — spawn
1 scheduler + S servers + W workers as OS processes on this host, wired
by DMLC_* env vars. Used by the CPU test-suite and bench.py.
"""

import multiprocessing as mp
import os
import random
import traceback


def _default_server_fn(ps, rank):
    server = ps.KVServer(0)
    server.set_default_handle()
    return None, server  # keep the server alive until finalize


def _child_main(role, rank, env, fn, args, q, device):
    try:
        os.environ.update(env)
        import ps_lite_amd as ps

        ps.init_env(env)
        ps.start(role=role, rank=rank, device=device)
        result = None
        keepalive = None
        if fn is not None:
            result = fn(ps, rank, *args)
            # convention: fn may return (payload, keepalive) where keepalive
            # holds live objects (e.g. a KVServer) until the finalize barrier
            if isinstance(result, tuple) and len(result) == 2:
                result, keepalive = result
        ps.finalize(role=role)
        del keepalive
        q.put((role, rank, "ok", result))
    except Exception:
        q.put((role, rank, "error", traceback.format_exc()))


class LocalCluster:
    """Spawn scheduler + servers + workers; collect worker results."""

    def __init__(self, num_workers, num_servers, env_extra=None, root_port=None,
                 joint=False, devices=None):
        self.num_workers = num_workers
        self.num_servers = num_servers
        self.joint = joint
        self.devices = devices or {}
        # stay BELOW the kernel's ephemeral range (default 32768+): a
        # transient outgoing socket from another process can otherwise
        # hold the scheduler's port and fail its bind
        port = root_port or random.randint(20000, 32000)
        self.env = {
            "DMLC_NUM_WORKER": str(num_workers),
            "DMLC_NUM_SERVER": str(num_servers),
            "DMLC_PS_ROOT_URI": "127.0.0.1",
            "DMLC_PS_ROOT_PORT": str(port),
        }
        if env_extra:
            self.env.update({k: str(v) for k, v in env_extra.items()})

    def run(self, worker_fn, server_fn=None, worker_args=(), timeout=120):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = []

        def spawn(role, rank, fn, args, device=-1):
            p = ctx.Process(target=_child_main, args=(role, rank, self.env, fn, args, q, device))
            p.daemon = True
            p.start()
            procs.append(p)

        spawn("scheduler", -1, None, ())
        if self.joint:
            for r in range(self.num_workers):
                spawn("joint", r, worker_fn, worker_args, self.devices.get(r, -1))
        else:
            sfn = server_fn or _default_server_fn
            for r in range(self.num_servers):
                spawn("server", r, sfn, (), self.devices.get(("server", r), -1))
            for r in range(self.num_workers):
                spawn("worker", r, worker_fn, worker_args, self.devices.get(("worker", r), -1))

        expected = len(procs)
        results = []
        for _ in range(expected):
            role, rank, status, payload = q.get(timeout=timeout)
            if status == "error":
                for p in procs:
                    p.terminate()
                raise RuntimeError(f"{role}:{rank} failed:\n{payload}")
            results.append((role, rank, payload))
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        out = {}
        for role, rank, payload in results:
            if role in ("worker", "joint"):
                out[rank] = payload
        return out


def launch_local(num_workers, num_servers, worker_fn, server_fn=None, env_extra=None,
                 joint=False, timeout=120, worker_args=(), devices=None):
    last = None
    for _ in range(3):  # a busy random port fails the scheduler bind: retry fresh
        c = LocalCluster(num_workers, num_servers, env_extra=env_extra, joint=joint,
                         devices=devices)
        try:
            return c.run(worker_fn, server_fn=server_fn, timeout=timeout,
                         worker_args=worker_args)
        except RuntimeError as e:
            if "failed to bind port" not in str(e):
                raise
            last = e
    raise last


def main():
    """CLI launcher (reference parity: tracker/dmlc_local.py): run an
    arbitrary app command as S servers + W workers + scheduler on this
    host, with the keepalive restart loop (retry a nonzero-exit process
    up to --retries times)."""
    import argparse
    import subprocess
    import sys
    import threading

    p = argparse.ArgumentParser(description=main.__doc__)
    p.add_argument("--workers", type=int, required=True)
    p.add_argument("--servers", type=int, required=True)
    p.add_argument("--root-port", type=int, default=9100)
    p.add_argument("--retries", type=int, default=3,
                   help="keepalive restarts per process (dmlc_local.py:15-23)")
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    a = p.parse_args()
    cmd = " ".join(c for c in a.cmd if c != "--")
    base = {
        "DMLC_NUM_WORKER": str(a.workers),
        "DMLC_NUM_SERVER": str(a.servers),
        "DMLC_PS_ROOT_URI": "127.0.0.1",
        "DMLC_PS_ROOT_PORT": str(a.root_port),
    }
    rcs = []

    def run(role, rank):
        env = dict(os.environ, **base, DMLC_ROLE=role)
        if rank >= 0:
            env["DMLC_RANK"] = str(rank)
        for _ in range(max(1, a.retries)):
            rc = subprocess.call(cmd, shell=True, env=env)
            if rc == 0:
                rcs.append(0)
                return
        rcs.append(rc)

    sched = subprocess.Popen(
        [sys.executable, "-c",
         "import ps_lite_amd as ps; ps.start(role='scheduler', device=-1); "
         "ps.finalize(role='scheduler')"],
        env=dict(os.environ, **base, DMLC_ROLE="scheduler"))
    threads = [threading.Thread(target=run, args=("server", r)) for r in range(a.servers)]
    threads += [threading.Thread(target=run, args=("worker", r)) for r in range(a.workers)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    sched.wait()
    sys.exit(0 if all(rc == 0 for rc in rcs) else 1)


if __name__ == "__main__":
    main()
