from .timing import Timer, human_bytes, human_rate  # noqa: F401
from .partition import partition_tensor, DEFAULT_PART_BYTES  # noqa: F401
