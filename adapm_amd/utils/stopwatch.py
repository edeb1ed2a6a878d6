"""Small timing utility (parity with reference include/utils.h Stopwatch,
:85-155)."""
import time


class Stopwatch:
    def __init__(self, running: bool = False):
        self._elapsed = 0.0
        self._start = time.perf_counter() if running else None

    def start(self):
        if self._start is None:
            self._start = time.perf_counter()
        return self

    def stop(self):
        if self._start is not None:
            self._elapsed += time.perf_counter() - self._start
            self._start = None
        return self

    def resume(self):
        return self.start()

    def elapsed(self) -> float:
        e = self._elapsed
        if self._start is not None:
            e += time.perf_counter() - self._start
        return e

    def __str__(self):
        return f"{self.elapsed():.3f}s"
