"""Kernel-only probe for rocprofv3 --pmc runs: launches the dense assign
and sum kernels on pool buffers with NO cluster (single process, no
sockets/threads), so counter collection sees clean dispatches."""
import sys
import pathlib

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent.parent))
import numpy as np
import ps_lite_amd as ps

NBYTES = 64 << 20
ITERS = 10

ps.pool_init(0, 2 * NBYTES + (64 << 20))
a = ps.pool_alloc(NBYTES)
b = ps.pool_alloc(NBYTES)
a.copy_from(np.random.default_rng(0).standard_normal(NBYTES // 4).astype(np.float32))
for _ in range(ITERS):
    ps._core.k_dense_assign(b.ptr, a.ptr, NBYTES)      # copy: 64 MiB r + 64 MiB w
for _ in range(ITERS):
    ps._core.k_dense_sum_f32(b.ptr, a.ptr, NBYTES // 4)  # sum: 128 MiB r + 64 MiB w
ps.device_sync(0)
print("PMC_PROBE_OK")
