"""Rate-limited work queues with client-go semantics.

The reference uses ``workqueue.NewNamedRateLimitingQueue(
workqueue.DefaultControllerRateLimiter(), name)`` (e.g.
``pkg/controller/globalaccelerator/controller.go:66-67``).  This module
reproduces the three layers of that stack:

- base queue: FIFO with *dirty* / *processing* sets so a key queued while
  being processed is re-queued exactly once after ``done()`` (coalescing);
- delaying queue: ``add_after`` delivers items after a delay via a single
  background timer thread per queue;
- rate limiting: per-item exponential backoff (client-go
  ``ItemExponentialFailureRateLimiter``, base 5ms, cap 1000s) combined with
  an overall token-bucket (client-go default 10 qps / burst 100).
"""

from __future__ import annotations

import heapq
import itertools
import threading
import time
from typing import Any, Optional, Tuple

from .. import metrics


class ItemExponentialFailureRateLimiter:
    """base * 2^failures per item, capped.  client-go defaults: 5ms..1000s."""

    def __init__(self, base_delay: float = 0.005, max_delay: float = 1000.0):
        self.base_delay = base_delay
        self.max_delay = max_delay
        self._failures = {}
        self._lock = threading.Lock()

    def when(self, item) -> float:
        with self._lock:
            failures = self._failures.get(item, 0)
            self._failures[item] = failures + 1
        delay = self.base_delay * (2**failures)
        return min(delay, self.max_delay)

    def forget(self, item):
        with self._lock:
            self._failures.pop(item, None)

    def num_requeues(self, item) -> int:
        with self._lock:
            return self._failures.get(item, 0)


class BucketRateLimiter:
    """Token bucket shared across items (client-go rate.Limiter(10, 100))."""

    def __init__(self, qps: float = 10.0, burst: int = 100):
        self.qps = qps
        self.burst = burst
        self._tokens = float(burst)
        self._last = time.monotonic()
        self._lock = threading.Lock()

    def when(self, item) -> float:
        with self._lock:
            now = time.monotonic()
            self._tokens = min(self.burst, self._tokens + (now - self._last) * self.qps)
            self._last = now
            self._tokens -= 1.0
            if self._tokens >= 0:
                return 0.0
            return -self._tokens / self.qps

    def forget(self, item):
        pass

    def num_requeues(self, item) -> int:
        return 0


class MaxOfRateLimiter:
    """client-go DefaultControllerRateLimiter: max of exponential + bucket."""

    def __init__(self, *limiters):
        self.limiters = limiters

    def when(self, item) -> float:
        return max(l.when(item) for l in self.limiters)

    def forget(self, item):
        for l in self.limiters:
            l.forget(item)

    def num_requeues(self, item) -> int:
        return max(l.num_requeues(item) for l in self.limiters)


def default_controller_rate_limiter() -> MaxOfRateLimiter:
    return MaxOfRateLimiter(
        ItemExponentialFailureRateLimiter(), BucketRateLimiter()
    )


class RateLimitingQueue:
    """Named rate-limiting work queue.

    API (snake_case of client-go RateLimitingInterface): ``add``,
    ``add_after``, ``add_rate_limited``, ``get`` -> (item, shutdown),
    ``done``, ``forget``, ``num_requeues``, ``shut_down``, ``len``.
    """

    def __init__(self, rate_limiter=None, name: str = ""):
        self.name = name
        self._rate_limiter = rate_limiter or default_controller_rate_limiter()
        self._cond = threading.Condition()
        self._queue: list = []
        self._dirty: set = set()
        self._processing: set = set()
        self._shutting_down = False
        # per-item timestamps for the client-go-style latency metrics
        self._added_at: dict = {}
        self._started_at: dict = {}
        # delayed delivery
        self._waiting: list = []  # heap of (ready_at, seq, item)
        self._seq = itertools.count()
        self._waiting_thread: Optional[threading.Thread] = None

    def _report_depth(self):
        metrics.set_queue_depth(self.name, len(self._queue))

    # -- base queue --------------------------------------------------------
    def add(self, item: Any):
        with self._cond:
            if self._shutting_down:
                return
            if item in self._dirty:
                return
            self._dirty.add(item)
            if item in self._processing:
                return
            self._queue.append(item)
            self._added_at.setdefault(item, time.monotonic())
            self._report_depth()
            self._cond.notify_all()

    def get(self, timeout: Optional[float] = None) -> Tuple[Any, bool]:
        """Blocks until an item is available or the queue shuts down.
        Returns (item, shutdown); shutdown=True means stop the worker."""
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._cond:
            while not self._queue and not self._shutting_down:
                remaining = None
                if deadline is not None:
                    remaining = deadline - time.monotonic()
                    if remaining <= 0:
                        return None, False
                self._cond.wait(remaining)
            if not self._queue:
                # shutting down and drained
                return None, True
            item = self._queue.pop(0)
            self._processing.add(item)
            self._dirty.discard(item)
            now = time.monotonic()
            added = self._added_at.pop(item, None)
            if added is not None:
                metrics.observe_queue_latency(self.name, now - added)
            self._started_at[item] = now
            self._report_depth()
            return item, False

    def done(self, item: Any):
        with self._cond:
            self._processing.discard(item)
            started = self._started_at.pop(item, None)
            if started is not None:
                metrics.observe_work_duration(self.name, time.monotonic() - started)
            if item in self._dirty:
                self._queue.append(item)
                self._added_at.setdefault(item, time.monotonic())
                self._cond.notify_all()

    def __len__(self) -> int:
        with self._cond:
            return len(self._queue)

    def shut_down(self):
        with self._cond:
            self._shutting_down = True
            self._cond.notify_all()

    def shutting_down(self) -> bool:
        with self._cond:
            return self._shutting_down

    # -- delaying queue ----------------------------------------------------
    def add_after(self, item: Any, delay: float):
        if delay <= 0:
            self.add(item)
            return
        with self._cond:
            if self._shutting_down:
                return
            heapq.heappush(self._waiting, (time.monotonic() + delay, next(self._seq), item))
            if self._waiting_thread is None:
                self._waiting_thread = threading.Thread(
                    target=self._waiting_loop, name=f"workqueue-{self.name}-delay", daemon=True
                )
                self._waiting_thread.start()
            self._cond.notify_all()

    def _waiting_loop(self):
        while True:
            with self._cond:
                if self._shutting_down:
                    return
                now = time.monotonic()
                while self._waiting and self._waiting[0][0] <= now:
                    _, _, item = heapq.heappop(self._waiting)
                    if item not in self._dirty:
                        self._dirty.add(item)
                        if item not in self._processing:
                            self._queue.append(item)
                            self._cond.notify_all()
                wait_for = self._waiting[0][0] - now if self._waiting else None
                self._cond.wait(timeout=wait_for)

    # -- rate limiting -----------------------------------------------------
    def add_rate_limited(self, item: Any):
        metrics.count_queue_retry(self.name)
        self.add_after(item, self._rate_limiter.when(item))

    def forget(self, item: Any):
        self._rate_limiter.forget(item)

    def num_requeues(self, item: Any) -> int:
        return self._rate_limiter.num_requeues(item)
