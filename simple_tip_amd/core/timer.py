"""Cumulative stopwatch used for the [setup, pred, quant, cam] timing
taxonomy (role of reference src/core/timer.py; written fresh — the reference
semantics kept are: cumulative across start/stop cycles, context-manager and
decorator forms, error on double start/stop, warn when read while running).

``DeviceTimer`` additionally synchronises the GPU on stop so asynchronously
launched kernels bill to the bucket that launched them; reads use
``time.perf_counter`` (monotonic), not wall-clock.
"""

import time
import warnings


class Timer:
    """Accumulates elapsed seconds over any number of start/stop cycles."""

    __slots__ = ("_total", "_t0")

    def __init__(self, start: bool = False):
        self._total = 0.0
        self._t0 = None  # perf_counter at the last start(), None when idle
        if start:
            self.start()

    @property
    def running(self) -> bool:
        return self._t0 is not None

    def start(self) -> "Timer":
        if self.running:
            raise RuntimeError("Timer is already started")
        self._t0 = time.perf_counter()
        return self

    def stop(self) -> float:
        """Stop and return the accumulated total."""
        if not self.running:
            raise RuntimeError("Timer is not started")
        self._total += time.perf_counter() - self._t0
        self._t0 = None
        return self._total

    def get(self) -> float:
        """Accumulated seconds; warns (and excludes the open interval) if
        the timer is still running."""
        if self.running:
            warnings.warn("Timer is not stopped", RuntimeWarning)
        return self._total

    def timed(self, fn):
        """Decorator form: bill every call of ``fn`` to this timer."""

        def timed_call(*args, **kwargs):
            self.start()
            try:
                return fn(*args, **kwargs)
            finally:
                self.stop()

        return timed_call

    def __enter__(self) -> "Timer":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()


class DeviceTimer(Timer):
    """A :class:`Timer` whose stop() first drains the GPU, so device work
    launched inside the interval is included in it. No-op on CPU-only."""

    def stop(self) -> float:
        try:
            import torch

            if torch.cuda.is_available():
                torch.cuda.synchronize()
        except Exception:  # pragma: no cover - torch is always importable here
            pass
        return super().stop()
