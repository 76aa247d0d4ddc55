from .timing import Timer, human_bytes, human_rate  # noqa: F401
