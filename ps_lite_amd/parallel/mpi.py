"""MPI cluster launcher.

Reference parity: ps-lite tracker/dmlc_mpi.py — start the scheduler
locally and spawn workers/servers via mpirun. Each MPI rank decides its
role from its rank index: ranks [0, num_servers) are servers, the rest
workers. Usage:

    python -m ps_lite_amd.parallel.mpi --workers 4 --servers 4 \
        --root-uri <ip> -- python my_app.py
"""

import argparse
import os
import subprocess
import sys


def launch_mpi(num_workers, num_servers, cmd, root_uri, root_port, hostfile=None,
               mpirun="mpirun"):
    base_env = {
        "DMLC_NUM_WORKER": str(num_workers),
        "DMLC_NUM_SERVER": str(num_servers),
        "DMLC_PS_ROOT_URI": root_uri,
        "DMLC_PS_ROOT_PORT": str(root_port),
    }
    sched_env = dict(os.environ, **base_env, DMLC_ROLE="scheduler")
    sched = subprocess.Popen(
        [sys.executable, "-c",
         "import ps_lite_amd as ps; ps.start(role='scheduler', device=-1); "
         "ps.finalize(role='scheduler')"],
        env=sched_env)
    # the rank->role shim runs inside each MPI process
    shim = (
        "import os, subprocess, sys;"
        "r = int(os.environ.get('OMPI_COMM_WORLD_RANK', os.environ.get('PMI_RANK', '0')));"
        f"role = 'server' if r < {num_servers} else 'worker';"
        f"rank = r if r < {num_servers} else r - {num_servers};"
        "os.environ['DMLC_ROLE'] = role; os.environ['DMLC_RANK'] = str(rank);"
        f"sys.exit(subprocess.call({cmd!r}, shell=True))"
    )
    margs = [mpirun, "-n", str(num_workers + num_servers)]
    if hostfile:
        margs += ["--hostfile", hostfile]
    for k, v in base_env.items():
        margs += ["-x", f"{k}={v}"]
    margs += [sys.executable, "-c", shim]
    rc = subprocess.call(margs, env=dict(os.environ, **base_env))
    sched.wait()
    return rc


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--workers", type=int, required=True)
    p.add_argument("--servers", type=int, required=True)
    p.add_argument("--root-uri", required=True)
    p.add_argument("--root-port", type=int, default=9100)
    p.add_argument("--hostfile", default=None)
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    a = p.parse_args()
    cmd = " ".join(c for c in a.cmd if c != "--")
    sys.exit(launch_mpi(a.workers, a.servers, cmd, a.root_uri, a.root_port, a.hostfile))


if __name__ == "__main__":
    main()
