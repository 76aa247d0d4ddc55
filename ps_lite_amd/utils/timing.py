"""Small timing/formatting helpers used by benchmarks and examples."""

import time


class Timer:
    """Context-manager wall timer: `with Timer() as t: ...; t.s`."""

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.s = time.perf_counter() - self.t0
        return False


def human_bytes(n):
    for unit in ("B", "KiB", "MiB", "GiB", "TiB"):
        if n < 1024 or unit == "TiB":
            return f"{n:.1f} {unit}"
        n /= 1024


def human_rate(bytes_per_s):
    return human_bytes(bytes_per_s) + "/s"
