"""Tensor partitioning (reference parity: BytePS splits big tensors into
sub-keys of <= BYTEPS_PARTITION_BYTES, 4 MB default — rdma_transport.h:593-596,
test_ipc_benchmark.cc:41-43. The partition keys spread round-robin across
the server key ranges so one tensor's parts move in parallel)."""
import numpy as np

DEFAULT_PART_BYTES = 4 << 20


def partition_tensor(nbytes, num_servers, base=0, part_bytes=DEFAULT_PART_BYTES,
                     elem_bytes=4):
    """Split a tensor of `nbytes` into <= part_bytes sub-keys.

    Returns (keys: sorted uint64 array, lens: int32 element counts, one
    per key, aligned with `keys`). Part j lands in server
    (j % num_servers)'s key range at slot `base + j // num_servers` —
    round-robin, so one tensor's parts use every server's bandwidth.
    """
    assert nbytes % elem_bytes == 0
    total_elems = nbytes // elem_bytes
    nparts = max(1, (nbytes + part_bytes - 1) // part_bytes)
    q, r = divmod(total_elems, nparts)
    step = (1 << 64) // num_servers
    pairs = []
    for j in range(nparts):
        key = (j % num_servers) * step + base + j // num_servers
        pairs.append((key, q + 1 if j < r else q))
    pairs.sort()
    keys = np.array([k for k, _ in pairs], dtype=np.uint64)
    lens = np.array([n for _, n in pairs], dtype=np.int32)
    return keys, lens
