"""Sparse embedding-table model for BASELINE config #5.

A rows x width fp32 table sharded over servers by key range: the global
row id r maps to key r << key_shift, so the even split of the 2^64 key
space (Postoffice::GetServerKeyRanges) shards rows evenly. Each server
holds rows/num_servers local rows (GpuSparseHandler with key_shift)."""

import numpy as np


class EmbeddingSpec:
    def __init__(self, rows=1 << 20, width=64):
        assert rows & (rows - 1) == 0, "rows must be a power of two"
        self.rows = rows
        self.width = width
        self.key_shift = 64 - rows.bit_length() + 1  # rows = 2^b -> shift = 64-b

    def rows_local(self, num_servers):
        assert self.rows % num_servers == 0
        return self.rows // num_servers

    def keys_for_rows(self, row_ids):
        rows = np.asarray(row_ids, dtype=np.uint64)
        return rows << np.uint64(self.key_shift)

    def hot_batch(self, nsel, seed=0):
        """Sorted unique random row ids (one training step's hot set)."""
        rng = np.random.default_rng(seed)
        rows = rng.choice(self.rows, size=nsel, replace=False)
        rows.sort()
        return self.keys_for_rows(rows)
