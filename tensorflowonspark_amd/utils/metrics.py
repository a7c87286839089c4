"""Step timing / throughput metrics (survey §5: the reference had none beyond
TensorBoard; a step-timer + images-per-sec logger is a north-star requirement).

``StepTimer`` logs rolling throughput; ``trace_span`` emits roctracer/rocprof
user markers when the extension is available (visible in rocprofv3 traces),
and is a no-op otherwise.
"""

import contextlib
import logging
import time

logger = logging.getLogger(__name__)


class StepTimer:
    """Rolling images/sec + ms/step logger.

    >>> timer = StepTimer(batch_size=256, log_every=10)
    >>> for batch in loader:
    ...     train_step(batch)
    ...     timer.step()
    """

    def __init__(self, batch_size, log_every=20, name="train", sync_fn=None):
        self.batch_size = batch_size
        self.log_every = log_every
        self.name = name
        self.sync_fn = sync_fn
        self.total_steps = 0
        self._t0 = None
        self._window_steps = 0

    def start(self):
        if self.sync_fn:
            self.sync_fn()
        self._t0 = time.time()
        self._window_steps = 0

    def step(self, n=1):
        if self._t0 is None:
            self.start()
            return None
        self.total_steps += n
        self._window_steps += n
        if self._window_steps >= self.log_every:
            if self.sync_fn:
                self.sync_fn()
            dt = time.time() - self._t0
            ips = self._window_steps * self.batch_size / dt
            ms = dt / self._window_steps * 1000
            logger.info("[%s] step %d: %.1f images/sec, %.2f ms/step",
                        self.name, self.total_steps, ips, ms)
            self._t0 = time.time()
            self._window_steps = 0
            self._last_rate = ips
            return ips
        return None

    def rate(self):
        """Most recent windowed images/sec (0.0 before the first window)."""
        return getattr(self, "_last_rate", 0.0)


@contextlib.contextmanager
def trace_span(name):
    """roctx-style span: shows up in rocprofv3 --marker-trace captures."""
    pushed = False
    try:
        try:
            import torch
            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(name)  # maps to roctx on ROCm
                pushed = True
        except Exception:
            pass
        yield
    finally:
        if pushed:
            try:
                import torch
                torch.cuda.nvtx.range_pop()
            except Exception:
                pass
