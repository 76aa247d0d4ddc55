"""Torch-tensor wrappers over the CDNA4 dense kernels (csrc/kernels.hip).

Synchronous convenience API for applications and tests; the PS server
handlers launch the same kernels asynchronously on per-peer streams.
"""

from .. import _core


def _check_f32_cuda(*tensors):
    for t in tensors:
        assert t.is_cuda, "expected a CUDA (HIP) tensor"
        assert t.dtype.itemsize == 4, "fp32 kernels"
        assert t.is_contiguous()


def dense_sum(dst, src):
    """dst += src (fp32, element-wise, float4-vectorized)."""
    _check_f32_cuda(dst, src)
    assert dst.numel() == src.numel()
    _core.k_dense_sum_f32(dst.data_ptr(), src.data_ptr(), dst.numel())
    return dst


def dense_assign(dst, src):
    """dst[:] = src (byte copy kernel; works across hipIpc mappings)."""
    _check_f32_cuda(dst, src)
    assert dst.numel() == src.numel()
    _core.k_dense_assign(dst.data_ptr(), src.data_ptr(), dst.numel() * dst.dtype.itemsize)
    return dst
