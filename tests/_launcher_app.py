"""Role-dispatching app used by the ssh/mpi launcher tests (the shape of
an application script under ps-lite's tracker/ launchers)."""
import os
import sys

import numpy as np

import ps_lite_amd as ps


def main():
    role = os.environ["DMLC_ROLE"]
    ps.start(role=role, device=-1)
    if role == "server":
        server = ps.KVServer(0)
        server.set_default_handle()
    elif role == "worker":
        w = ps.KVWorker(0, 0)
        keys = np.array([7], dtype=np.uint64)
        vals = np.ones(64, dtype=np.float32)
        w.wait(w.push(keys, vals, np.array([64], dtype=np.int32)))
        out = w.pull(keys)
        assert float(out.sum()) == 64.0, out
        print("WORKER_OK")
    ps.finalize(role=role)
    sys.exit(0)


if __name__ == "__main__":
    main()
