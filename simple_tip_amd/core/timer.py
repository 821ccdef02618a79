"""Cumulative wall-clock timer (capability parity with reference
src/core/timer.py:6-50), plus a GPU-aware variant that synchronises the
device so kernel time is attributed to the right bucket."""

import time
import warnings


class Timer:
    """Cumulative wall-clock timer usable as context manager or decorator."""

    def __init__(self, start: bool = False):
        self._start_time = None
        self._elapsed = 0.0
        if start:
            self.start()

    def start(self):
        """Start the timer; it must not already be running."""
        if self._start_time is not None:
            raise RuntimeError("Timer is already started")
        self._start_time = time.time()

    def stop(self):
        """Stop the timer; it must be running."""
        if self._start_time is None:
            raise RuntimeError("Timer is not started")
        self._elapsed += time.time() - self._start_time
        self._start_time = None

    def timed(self, f):
        """Decorator: accumulate the wrapped call's wall time into this timer."""

        def wrapper(*args, **kwargs):
            with self:
                return f(*args, **kwargs)

        return wrapper

    def get(self) -> float:
        """Elapsed seconds. Warns if the timer is still running."""
        if self._start_time is not None:
            warnings.warn("Timer is not stopped", RuntimeWarning)
        return self._elapsed

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()


class DeviceTimer(Timer):
    """Like :class:`Timer` but synchronises the CUDA/HIP device on stop so
    asynchronously launched kernels are billed to the bucket that launched
    them. No-op without a GPU."""

    def stop(self):
        try:
            import torch

            if torch.cuda.is_available():
                torch.cuda.synchronize()
        except Exception:  # pragma: no cover - torch always importable here
            pass
        super().stop()
