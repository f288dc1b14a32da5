"""Rate-limited, deduplicating work queue (client-go workqueue analog).

Controller-runtime serializes reconciles per key through exactly this
structure; the semantics preserved here: an item re-added while queued is
coalesced; an item re-added while being processed is re-queued after the
current reconcile finishes (dirty set); failed items back off exponentially.
"""
from __future__ import annotations

import heapq
import threading
import time
from typing import Dict, Hashable, List, Optional, Set, Tuple


class RateLimitingQueue:
    def __init__(self, base_delay: float = 0.005, max_delay: float = 16.0):
        self._lock = threading.Condition()
        self._queue: List[Hashable] = []
        self._queued: Set[Hashable] = set()
        self._processing: Set[Hashable] = set()
        self._dirty: Set[Hashable] = set()
        self._delayed: List[Tuple[float, int, Hashable]] = []  # heap
        # earliest pending deadline per item: a key requeued every reconcile
        # (requeue_after) must hold ONE live heap entry, not one per
        # reconcile — an unbounded heap was the main churn-RSS growth in the
        # round-1 soak (~165k stale entries at 550 reconciles/s × 300 s)
        self._delayed_next: Dict[Hashable, float] = {}
        self._failures: Dict[Hashable, int] = {}
        self._seq = 0
        self._shutdown = False
        self.base_delay = base_delay
        self.max_delay = max_delay

    def add(self, item: Hashable) -> None:
        with self._lock:
            if self._shutdown:
                return
            if item in self._processing:
                self._dirty.add(item)
                return
            if item in self._queued:
                return
            self._queued.add(item)
            self._queue.append(item)
            self._lock.notify()

    def add_after(self, item: Hashable, delay: float) -> None:
        if delay <= 0:
            self.add(item)
            return
        with self._lock:
            if self._shutdown:
                return
            when = time.monotonic() + delay
            pending = self._delayed_next.get(item)
            if pending is not None and pending <= when:
                return  # already scheduled at least as soon; entry coalesced
            self._delayed_next[item] = when
            self._seq += 1
            heapq.heappush(self._delayed, (when, self._seq, item))
            self._lock.notify()

    def add_rate_limited(self, item: Hashable) -> None:
        with self._lock:
            failures = self._failures.get(item, 0)
            self._failures[item] = failures + 1
        delay = min(self.base_delay * (2 ** failures), self.max_delay)
        self.add_after(item, delay)

    def forget(self, item: Hashable) -> None:
        with self._lock:
            self._failures.pop(item, None)

    def _drain_delayed(self) -> float:
        """Move due delayed items to the main queue. Returns wait hint."""
        now = time.monotonic()
        wait = 3600.0
        while self._delayed:
            when, _, item = self._delayed[0]
            if when <= now:
                heapq.heappop(self._delayed)
                # stale entry: a sooner deadline was registered after this
                # one was pushed (add_after coalescing) and already fired
                if self._delayed_next.get(item) != when:
                    continue
                del self._delayed_next[item]
                if item not in self._queued and item not in self._processing:
                    self._queued.add(item)
                    self._queue.append(item)
                elif item in self._processing:
                    self._dirty.add(item)
            else:
                wait = min(wait, when - now)
                break
        return wait

    def get(self, timeout: Optional[float] = None) -> Optional[Hashable]:
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._lock:
            while True:
                wait = self._drain_delayed()
                if self._queue:
                    item = self._queue.pop(0)
                    self._queued.discard(item)
                    self._processing.add(item)
                    return item
                if self._shutdown:
                    return None
                if deadline is not None:
                    remaining = deadline - time.monotonic()
                    if remaining <= 0:
                        return None
                    wait = min(wait, remaining)
                self._lock.wait(wait)

    def done(self, item: Hashable) -> None:
        with self._lock:
            self._processing.discard(item)
            if item in self._dirty:
                self._dirty.discard(item)
                if item not in self._queued:
                    self._queued.add(item)
                    self._queue.append(item)
                    self._lock.notify()

    def shutdown(self) -> None:
        with self._lock:
            self._shutdown = True
            self._lock.notify_all()

    def __len__(self) -> int:
        with self._lock:
            return len(self._queue) + len(self._delayed_next)
