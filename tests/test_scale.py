"""Scale-shaped CPU regressions.

1. An 8-worker + 8-server joint bring-up over the host shm plane — the
   same 17-node bootstrap (ADD_NODE batch, staged import barrier ladder,
   8 inbound rings) the driver's 8-GPU scaling bench goes through, minus
   the GPUs (reference shape: tests/local.sh N-process localhost pattern,
   SURVEY.md §4).
2. Wire-format robustness: truncated / corrupt frames must die loudly in
   UnpackMeta's bounds checks (never read out of bounds), exercised from
   a subprocess since XPS_CHECK aborts.
"""
import os
import subprocess
import sys

import numpy as np

from ps_lite_amd.parallel import launch_local

N = 8
KEYS_PER_WORKER = 4
VLEN = 1024  # floats per key


def _scale_worker(ps_mod, rank):
    server = ps_mod.KVServer(0)
    server.set_default_handle()
    w = ps_mod.KVWorker(0, 0)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    rng = np.random.default_rng(77 + rank)
    # keys spread over every server's range so each worker talks to all 8
    step = (1 << 64) // N
    keys = np.array(sorted((s * step) + 1 + rank for s in range(N)), dtype=np.uint64)
    lens = np.full(len(keys), VLEN, dtype=np.int32)
    vals = rng.standard_normal(len(keys) * VLEN).astype(np.float32)
    for it in range(1, 4):  # the default handle accumulates pushes
        w.wait(w.push(keys, vals, lens))
        got = w.pull(keys)
        assert np.allclose(got, it * vals, atol=1e-5), \
            "pulled values mismatch at rank %d iter %d" % (rank, it)
    ps_mod.barrier("worker", ps_mod.WORKER_GROUP)
    return float(got.sum())


def test_scale_8x8_joint():
    results = launch_local(N, N, _scale_worker, joint=True, timeout=420)
    assert len(results) == N
    for r, v in results.items():
        assert np.isfinite(v)


_FUZZ_SNIPPET = r"""
import sys
import ps_lite_amd as ps
buf = ps._core._pack_meta_sample()
mode = sys.argv[1]
if mode == "roundtrip":
    assert ps._core._unpack_meta_raw(buf)
    print("OK")
elif mode == "truncate":
    ps._core._unpack_meta_raw(buf[: int(sys.argv[2])])
elif mode == "garbage":
    import os
    b = bytearray(buf)
    b[0] = 1  # keep the version byte valid, corrupt the rest
    for i in range(1, len(b)):
        b[i] = (b[i] + 0x9D) & 0xFF
    ps._core._unpack_meta_raw(bytes(b))
"""


def _run_fuzz(*args):
    env = dict(os.environ)
    env.setdefault("PYTHONPATH", os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    return subprocess.run([sys.executable, "-c", _FUZZ_SNIPPET, *args],
                          capture_output=True, text=True, timeout=120, env=env)


def test_wire_roundtrip_raw():
    p = _run_fuzz("roundtrip")
    assert p.returncode == 0 and "OK" in p.stdout, p.stderr


def test_wire_truncated_frames_abort():
    full = None
    p = _run_fuzz("roundtrip")
    assert p.returncode == 0
    # every strict prefix must be rejected by a bounds CHECK, not UB
    for cut in (1, 5, 9, 17, 33, 49):
        p = _run_fuzz("truncate", str(cut))
        assert p.returncode != 0, f"truncated frame at {cut} bytes was accepted"
        assert "wire:" in p.stderr or "Check" in p.stderr or p.returncode < 0, p.stderr


def test_wire_garbage_frame_dies_loudly():
    p = _run_fuzz("garbage")
    # corrupt lengths/counts must hit a CHECK (nonzero exit), never hang
    # or silently succeed
    assert p.returncode != 0
