from .resnet50_buckets import resnet50_grad_buckets, resnet50_param_sizes  # noqa: F401
from .embedding import EmbeddingSpec  # noqa: F401
