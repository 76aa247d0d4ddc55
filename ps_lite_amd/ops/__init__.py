from .dense import dense_assign, dense_sum  # noqa: F401
from .sparse import sparse_gather, sparse_scatter_add  # noqa: F401
