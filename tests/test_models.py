"""Sanity checks for the benchmark model helpers."""
import numpy as np

from ps_lite_amd.models import EmbeddingSpec, resnet50_grad_buckets
from ps_lite_amd.models.resnet50_buckets import resnet50_param_sizes


def test_resnet50_geometry():
    sizes = resnet50_param_sizes()
    total = sum(sizes)
    # ResNet-50 (v1.5, 1000 classes) has ~25.56 M parameters
    assert 25_400_000 < total < 25_700_000, total
    buckets = resnet50_grad_buckets()
    assert sum(buckets) == total * 4  # bytes
    assert all(b <= 4 << 20 for b in buckets)  # BYTEPS_PARTITION_BYTES
    assert len(buckets) >= 100  # bucketized, not one blob


def test_embedding_spec_key_mapping():
    spec = EmbeddingSpec(rows=1 << 20, width=64)
    # key = row << shift must land row r in server (r * n // rows) under
    # the even key-range split
    n = 4
    step = (1 << 64) // n
    rows = np.array([0, (1 << 20) // 4, (1 << 20) // 2, (1 << 20) - 1], dtype=np.uint64)
    keys = spec.keys_for_rows(rows)
    servers = (keys // np.uint64(step)).astype(np.int64)
    assert servers.tolist() == [0, 1, 2, 3]
    assert spec.rows_local(n) * n >= 1 << 20
    # hot_batch: sorted unique
    k = spec.hot_batch(1000, seed=3)
    assert len(np.unique(k)) == 1000
    assert np.all(np.diff(k.astype(np.uint64)) > 0)
