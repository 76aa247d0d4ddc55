"""SSH cluster launcher.

Reference parity: ps-lite tracker/dmlc_ssh.py — start the scheduler
locally and workers/servers on remote hosts over ssh, wired by DMLC_*
env vars. Remote hosts need this repo at the same path (or pass
--repo). Includes the keepalive restart loop of tracker/dmlc_local.py
(retry on nonzero exit) which, combined with the Van's recovery path,
gives restart-based fault tolerance.
"""

import argparse
import os
import subprocess
import sys
import threading
import time


def _env_str(env):
    return " ".join(f"{k}={v}" for k, v in env.items())


def launch_ssh(hosts_workers, hosts_servers, cmd, root_uri, root_port,
               repo=None, retries=3, ssh_opts=()):
    """Launch `cmd` (a python script using ps_lite_amd) on remote hosts.

    hosts_workers/hosts_servers: list of "host" strings (ssh targets).
    Returns when every remote process has exited.
    """
    repo = repo or os.getcwd()
    base_env = {
        "DMLC_NUM_WORKER": str(len(hosts_workers)),
        "DMLC_NUM_SERVER": str(len(hosts_servers)),
        "DMLC_PS_ROOT_URI": root_uri,
        "DMLC_PS_ROOT_PORT": str(root_port),
    }
    procs = []

    def run_remote(host, role, rank):
        env = dict(base_env, DMLC_ROLE=role, DMLC_RANK=str(rank))
        remote = f"cd {repo} && {_env_str(env)} {cmd}"
        for attempt in range(retries):
            p = subprocess.Popen(["ssh", "-o", "StrictHostKeyChecking=no", *ssh_opts,
                                  host, remote])
            procs.append(p)
            p.wait()
            if p.returncode == 0:
                return
            time.sleep(2)  # keepalive restart (dmlc_local.py:15-23 behavior)

    sched_env = dict(base_env, DMLC_ROLE="scheduler")
    sched = subprocess.Popen([sys.executable, "-c",
                              "import ps_lite_amd as ps; ps.start(role='scheduler', device=-1); "
                              "ps.finalize(role='scheduler')"],
                             env={**os.environ, **sched_env}, cwd=repo)
    threads = []
    for r, h in enumerate(hosts_servers):
        t = threading.Thread(target=run_remote, args=(h, "server", r))
        t.start()
        threads.append(t)
    for r, h in enumerate(hosts_workers):
        t = threading.Thread(target=run_remote, args=(h, "worker", r))
        t.start()
        threads.append(t)
    for t in threads:
        t.join()
    sched.wait()


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--workers", required=True, help="comma-separated worker hosts")
    p.add_argument("--servers", required=True, help="comma-separated server hosts")
    p.add_argument("--root-uri", required=True)
    p.add_argument("--root-port", type=int, default=9100)
    p.add_argument("--repo", default=None)
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    a = p.parse_args()
    launch_ssh(a.workers.split(","), a.servers.split(","), " ".join(a.cmd),
               a.root_uri, a.root_port, repo=a.repo)


if __name__ == "__main__":
    main()
