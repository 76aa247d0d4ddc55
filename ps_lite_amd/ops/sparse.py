"""Torch-tensor wrappers over the CDNA4 sparse row kernels."""

from .. import _core


def sparse_gather(table, rows, out=None):
    """out[i] = table[rows[i]] for a 2-D fp32 table; rows int64/uint64."""
    import torch

    assert table.is_cuda and table.dim() == 2 and table.is_contiguous()
    assert rows.is_cuda and rows.dim() == 1
    if out is None:
        out = torch.empty(rows.numel(), table.shape[1], device=table.device,
                          dtype=table.dtype)
    _core.k_sparse_gather_f32(table.data_ptr(), rows.data_ptr(), rows.numel(),
                              table.shape[1], out.data_ptr())
    return out


def sparse_scatter_add(table, rows, src, atomic=False):
    """table[rows[i]] += src[i]; atomic=True tolerates duplicate rows."""
    assert table.is_cuda and table.dim() == 2 and table.is_contiguous()
    assert src.shape == (rows.numel(), table.shape[1])
    _core.k_sparse_scatter_add_f32(table.data_ptr(), rows.data_ptr(), rows.numel(),
                                   table.shape[1], src.data_ptr(), atomic)
    return table
