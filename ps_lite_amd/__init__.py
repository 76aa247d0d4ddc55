"""ps_lite_amd — MI355X-native parameter-server communication framework.

A from-scratch rebuild of the capabilities of bytedance/ps-lite (byteps
branch) designed for AMD Instinct MI355X: KVWorker/KVServer ZPush/ZPull
over a worker/server/scheduler process model, with device payloads in a
pre-registered HBM pool, a TCP side channel for control, and a same-host
data plane on shm rings + hipIpc/xGMI (see docs/DESIGN.md).
"""

import os

# Load torch's bundled HIP runtime FIRST when torch is present: _core
# links libamdhip64.so.7 and must bind to the same runtime instance as
# torch, else whichever loads second fails to initialize the GPU.
try:  # pragma: no cover
    import torch  # noqa: F401
except ImportError:
    pass

# the extension must be the in-tree build (fails loudly if missing)
try:
    from . import _core
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "ps_lite_amd._core extension not built. Run `make` at the repo root "
        "(hipcc --offload-arch=gfx950)."
    ) from e

from ._core import (  # noqa: F401
    KVWorker,
    KVServer,
    SimpleApp,
    PoolBuffer,
    SCHEDULER_GROUP,
    SERVER_GROUP,
    WORKER_GROUP,
    barrier,
    clear_registry,
    device_sync,
    finalize,
    gpu_count,
    host_alloc,
    init_env,
    my_rank,
    node_id,
    num_servers,
    num_workers,
    plane_peer_bytes,
    pool_alloc,
    pool_in_use,
    pool_init,
    recv_bytes,
    send_bytes,
    start,
)

__version__ = "0.1.0"


class _CudaArrayView:
    """Exports a PoolBuffer as __cuda_array_interface__ so torch can wrap it."""

    def __init__(self, buf, shape, typestr, itemsize):
        self._buf = buf
        self.__cuda_array_interface__ = {
            "shape": tuple(shape),
            "typestr": typestr,
            "data": (buf.ptr, False),
            "version": 2,
            "strides": None,
        }


def pool_tensor(shape, dtype="float32"):
    """A torch CUDA tensor backed by HBM-pool memory — ZPush/ZPull on its
    data_ptr() ride the zero-copy hipIpc fast path (pool membership is
    what the plane checks). The pool allocation is pinned to the tensor.
    """
    import numpy as np
    import torch

    shape = (shape,) if isinstance(shape, int) else tuple(shape)
    np_dtype = np.dtype(dtype)
    nbytes = int(np.prod(shape)) * np_dtype.itemsize
    buf = pool_alloc(nbytes)
    view = _CudaArrayView(buf, shape, np_dtype.str, np_dtype.itemsize)
    t = torch.as_tensor(view, device="cuda")
    t._xps_pool_buf = buf  # keep the pool allocation alive with the tensor
    return t


def use_torch_pool_allocator():
    """Route ALL torch CUDA allocations through the HbmPool (torch
    pluggable allocator), so ordinary ``torch.empty(..., device='cuda')``
    tensors live in the pool and ride the zero-copy hipIpc plane — no
    per-buffer registration, no ``pool_tensor`` wrapper (the reference's
    PinMemory / RegisterRecvBuffer use case, ucx_van.h:603-623).

    Must be called BEFORE the first CUDA allocation in the process
    (torch refuses to swap allocators afterwards).
    """
    import torch

    alloc = torch.cuda.memory.CUDAPluggableAllocator(
        _core.__file__, "xps_torch_alloc", "xps_torch_free")
    torch.cuda.memory.change_current_allocator(alloc)


def setup_env(num_workers, num_servers, root_uri="127.0.0.1", root_port=9100, **extra):
    """Set the DMLC_* environment both for this process and for children."""
    env = {
        "DMLC_NUM_WORKER": str(num_workers),
        "DMLC_NUM_SERVER": str(num_servers),
        "DMLC_PS_ROOT_URI": root_uri,
        "DMLC_PS_ROOT_PORT": str(root_port),
        # reset knobs that may linger from a previous cluster in this process
        "DMLC_GROUP_SIZE": "1",
        "DMLC_RANK": "-1",
    }
    env.update({k: str(v) for k, v in extra.items()})
    os.environ.update(env)
    init_env(env)
    return env
