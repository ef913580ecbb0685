"""Keyed, rate-limited work queue with "latest enqueue wins" semantics.

Functional equivalent of the reference's ``pkg/workqueue/workqueue.go:31-197``
and ``pkg/workqueue/jitterlimiter.go:31-66``:

* items are keyed; re-enqueueing a key supersedes the previously queued work
  for that key (latest wins),
* failed items are retried with a per-item exponential backoff combined with
  a global token bucket,
* an optional multiplicative jitter de-synchronizes retry storms across many
  daemons,
* per-item retry state resets on success (``Forget``).

Three limiter presets mirror the reference:
``prepare_unprepare_limiter`` (250 ms→3 s per-item expo + global 5 rps/burst
10, ``workqueue.go:49-59``), ``cd_daemon_limiter`` (5 ms→6 s expo with ±25 %
jitter, ``workqueue.go:61-63``), and ``default_controller_limiter``.
"""

from __future__ import annotations

import heapq
import random
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, Optional


class RateLimiter:
    """Per-item exponential backoff combined with a global token bucket."""

    def __init__(
        self,
        base_delay: float = 0.005,
        max_delay: float = 1000.0,
        qps: Optional[float] = None,
        burst: int = 10,
        jitter: float = 0.0,
    ):
        self._base = base_delay
        self._max = max_delay
        self._qps = qps
        self._burst = burst
        self._jitter = jitter
        self._failures: Dict[str, int] = {}
        self._lock = threading.Lock()
        # Reservation-style token bucket (client-go flowcontrol semantics):
        # requests beyond the burst reserve successive 1/qps slots, so a storm
        # of retries is spaced at the QPS rate instead of firing together.
        self._next_free = time.monotonic() - (burst / qps if qps else 0.0)

    def when(self, key: str) -> float:
        """Seconds to wait before retrying `key`."""
        with self._lock:
            n = self._failures.get(key, 0)
            self._failures[key] = n + 1
            delay = min(self._base * (2**n), self._max)
            if self._jitter:
                delay *= 1.0 + random.uniform(-self._jitter, self._jitter)
            if self._qps is not None:
                now = time.monotonic()
                interval = 1.0 / self._qps
                # This request's slot: either the next free slot, or now if
                # burst capacity remains (a full bucket covers burst slots).
                slot = max(self._next_free, now - (self._burst - 1) * interval)
                self._next_free = slot + interval
                delay = max(delay, slot - now)
            return delay

    def forget(self, key: str) -> None:
        with self._lock:
            self._failures.pop(key, None)

    def retries(self, key: str) -> int:
        with self._lock:
            return self._failures.get(key, 0)


def prepare_unprepare_limiter() -> RateLimiter:
    return RateLimiter(base_delay=0.25, max_delay=3.0, qps=5.0, burst=10)


def cd_daemon_limiter() -> RateLimiter:
    return RateLimiter(base_delay=0.005, max_delay=6.0, jitter=0.25)


def default_controller_limiter() -> RateLimiter:
    return RateLimiter(base_delay=0.005, max_delay=1000.0, qps=10.0, burst=100)


@dataclass(order=True)
class _Scheduled:
    ready_at: float
    seq: int
    key: str = field(compare=False)
    work: Callable[[], None] = field(compare=False)


class WorkQueue:
    """Threaded work queue: keyed supersede + rate-limited retries.

    ``enqueue(key, fn)`` schedules ``fn`` to run; if work for ``key`` is
    already queued (not yet running), the new ``fn`` replaces it.  If ``fn``
    raises, it is re-enqueued after ``limiter.when(key)`` — unless a newer
    enqueue for the key superseded it in the meantime.
    """

    def __init__(self, limiter: Optional[RateLimiter] = None, workers: int = 1, name: str = "wq"):
        self._limiter = limiter or default_controller_limiter()
        self._name = name
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._heap: list[_Scheduled] = []
        self._latest: Dict[str, int] = {}  # key -> newest seq
        self._seq = 0
        self._shutdown = False
        self._idle_cond = threading.Condition(self._lock)
        self._inflight = 0
        # Keys currently executing. With workers>1 a re-enqueued key must NOT
        # run concurrently with its still-in-flight predecessor — client-go's
        # workqueue guarantees per-key serialization via its dirty/processing
        # sets; we defer dispatch of a running key until the run finishes.
        self._running: set = set()
        self._threads = [
            threading.Thread(target=self._run, name=f"{name}-{i}", daemon=True)
            for i in range(workers)
        ]
        for t in self._threads:
            t.start()

    def enqueue(self, key: str, work: Callable[[], None], delay: float = 0.0) -> None:
        with self._lock:
            if self._shutdown:
                return
            self._seq += 1
            self._latest[key] = self._seq
            heapq.heappush(self._heap, _Scheduled(time.monotonic() + delay, self._seq, key, work))
            self._cond.notify()

    def forget(self, key: str) -> None:
        self._limiter.forget(key)

    def _run(self) -> None:
        while True:
            with self._lock:
                item = None
                while True:
                    if self._shutdown:
                        return
                    now = time.monotonic()
                    # Drop superseded entries lazily.
                    while self._heap and self._latest.get(self._heap[0].key) != self._heap[0].seq:
                        heapq.heappop(self._heap)
                    # Earliest ready item whose key is not in flight; ready
                    # items for running keys are deferred (per-key
                    # serialization) and re-pushed.
                    deferred = []
                    while self._heap and self._heap[0].ready_at <= now:
                        cand = heapq.heappop(self._heap)
                        if self._latest.get(cand.key) != cand.seq:
                            continue  # superseded
                        if cand.key in self._running:
                            deferred.append(cand)
                            continue
                        item = cand
                        break
                    for b in deferred:
                        heapq.heappush(self._heap, b)
                    if item is not None:
                        self._latest.pop(item.key, None)
                        self._running.add(item.key)
                        self._inflight += 1
                        break
                    timeout = (self._heap[0].ready_at - now) if self._heap else None
                    if timeout is not None and timeout <= 0:
                        # Head is ready but its key is running: wait for the
                        # completion notify, not a timed spin.
                        timeout = None
                    self._cond.wait(timeout=timeout)
            try:
                item.work()
                self._limiter.forget(item.key)
            except Exception:
                delay = self._limiter.when(item.key)
                with self._lock:
                    if not self._shutdown and item.key not in self._latest:
                        self._seq += 1
                        self._latest[item.key] = self._seq
                        heapq.heappush(
                            self._heap,
                            _Scheduled(time.monotonic() + delay, self._seq, item.key, item.work),
                        )
                        self._cond.notify()
            finally:
                with self._lock:
                    self._running.discard(item.key)
                    self._inflight -= 1
                    # Wake workers: a deferred same-key item may now be
                    # dispatchable.
                    self._cond.notify_all()
                    self._idle_cond.notify_all()

    def wait_idle(self, timeout: float = 10.0) -> bool:
        """Block until nothing is queued or running. Test helper."""
        deadline = time.monotonic() + timeout
        with self._lock:
            while self._heap or self._inflight:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    return False
                self._idle_cond.wait(timeout=min(remaining, 0.05))
            return True

    def shutdown(self) -> None:
        with self._lock:
            self._shutdown = True
            self._cond.notify_all()
        for t in self._threads:
            t.join(timeout=2.0)
