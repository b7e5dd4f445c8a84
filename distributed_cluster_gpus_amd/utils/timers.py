"""Wall-clock throughput meters.

The reference has no profiling hooks; events/sec is this
framework's headline metric, so every engine reports through
ThroughputMeter.
"""
import time


class ThroughputMeter:
    def __init__(self):
        self.t0 = None
        self.t1 = None
        self.count = 0

    def start(self):
        self.t0 = time.perf_counter()
        self.count = 0
        return self

    def add(self, n: int = 1):
        self.count += n

    def stop(self):
        self.t1 = time.perf_counter()
        return self

    @property
    def elapsed_s(self) -> float:
        end = self.t1 if self.t1 is not None else time.perf_counter()
        return max(1e-12, end - (self.t0 or end))

    @property
    def per_sec(self) -> float:
        return self.count / self.elapsed_s
