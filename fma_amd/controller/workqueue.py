"""Generic typed work-queue framework (reference pkg/controller/generic/
queue-work.go:35-141 and knows-processed-sync.go:34-103).

Semantics preserved from the reference:
- rate-limited re-queue with exponential per-item backoff capped at 20 s;
- an item added while being processed is re-processed afterwards (dirty
  set), never processed concurrently with itself;
- KnowsProcessedSync: sentinel items mark the initial batch; a callback
  fires once every item enqueued before start has been processed (the
  populator uses this to gate its key workers —
  reference populator.go:356-363).
"""

from __future__ import annotations

import threading
import time
from typing import Callable, Generic, Hashable, List, Optional, TypeVar

T = TypeVar("T", bound=Hashable)

MAX_BACKOFF_SECONDS = 20.0
BASE_BACKOFF_SECONDS = 0.005


class RateLimitingQueue(Generic[T]):
    """Subset of client-go's workqueue: Add / AddAfter / AddRateLimited /
    Forget / Get / Done / ShutDown with dirty/processing sets."""

    def __init__(self, max_backoff: float = MAX_BACKOFF_SECONDS,
                 metrics_name: str = "") -> None:
        self.max_backoff = max_backoff
        self._cond = threading.Condition()
        self._queue: List[T] = []
        self._dirty: set = set()
        self._processing: set = set()
        self._failures: dict = {}
        self._shutdown = False
        self._timers: List[threading.Timer] = []
        # the reference exports k8s workqueue metrics for its inner
        # queues (docs/metrics.md fma_dpc_innerqueue_*); emit the same
        # family when a name is given
        self._mname = metrics_name
        self._enqueued_at: dict = {}

    def add(self, item: T) -> None:
        with self._cond:
            if self._shutdown or item in self._dirty:
                return
            self._dirty.add(item)
            if self._mname:
                from fma_amd.controller import metrics as _m
                _m.queue_adds_total().labels(self._mname).inc()
                self._enqueued_at.setdefault(item, time.monotonic())
            if item in self._processing:
                return
            self._queue.append(item)
            if self._mname:
                from fma_amd.controller import metrics as _m
                _m.queue_depth().labels(self._mname).set(len(self._queue))
            self._cond.notify()

    def add_after(self, item: T, delay: float) -> None:
        if delay <= 0:
            self.add(item)
            return
        t = threading.Timer(delay, self.add, args=(item,))
        t.daemon = True
        with self._cond:
            if self._shutdown:
                return
            # prune fired timers so a long-running controller's retry
            # traffic does not grow this list without bound
            if len(self._timers) > 64:
                self._timers = [x for x in self._timers if x.is_alive()]
            self._timers.append(t)
        t.start()

    def add_rate_limited(self, item: T) -> None:
        with self._cond:
            n = self._failures.get(item, 0)
            self._failures[item] = n + 1
        if self._mname:
            from fma_amd.controller import metrics as _m
            _m.queue_retries_total().labels(self._mname).inc()
        self.add_after(item, min(BASE_BACKOFF_SECONDS * (2 ** n),
                                 self.max_backoff))

    def forget(self, item: T) -> None:
        with self._cond:
            self._failures.pop(item, None)

    def get(self) -> Optional[T]:
        with self._cond:
            while not self._queue and not self._shutdown:
                self._cond.wait(timeout=0.2)
            if not self._queue:
                return None
            item = self._queue.pop(0)
            self._dirty.discard(item)
            self._processing.add(item)
            if self._mname:
                from fma_amd.controller import metrics as _m
                _m.queue_depth().labels(self._mname).set(len(self._queue))
                t0 = self._enqueued_at.pop(item, None)
                if t0 is not None:
                    _m.queue_queue_duration_seconds().labels(
                        self._mname).observe(time.monotonic() - t0)
            return item

    def done(self, item: T) -> None:
        with self._cond:
            self._processing.discard(item)
            if item in self._dirty:
                self._queue.append(item)
                self._cond.notify()

    def shut_down(self) -> None:
        with self._cond:
            self._shutdown = True
            for t in self._timers:
                t.cancel()
            self._cond.notify_all()

    def __len__(self) -> int:
        with self._cond:
            return len(self._queue)


class QueueAndWorkers(Generic[T]):
    """Queue + N worker threads running `process(item)`.

    process returns False (done), True (retry with exponential backoff) or
    a float (re-queue after exactly that many seconds — the reference's
    processResult.retryAfter for known wait states, e.g. a server that is
    still booting; inference-server.go:448-452, :512)."""

    def __init__(self, name: str, num_workers: int,
                 process: Callable[[T], bool],
                 max_backoff: float = MAX_BACKOFF_SECONDS,
                 metrics_name: str = ""):
        self.name = name
        self.queue: RateLimitingQueue[T] = RateLimitingQueue(
            max_backoff, metrics_name=metrics_name)
        self._mname = metrics_name
        self.num_workers = num_workers
        self.process = process
        self.threads: List[threading.Thread] = []

    def start(self) -> None:
        for i in range(self.num_workers):
            th = threading.Thread(target=self._worker, daemon=True,
                                  name=f"{self.name}-worker-{i}")
            th.start()
            self.threads.append(th)

    def _worker(self) -> None:
        while True:
            item = self.queue.get()
            if item is None:
                return
            t0 = time.monotonic()
            try:
                retry = self.process(item)
            except Exception:  # noqa: BLE001 - reconcile must not kill worker
                import traceback
                traceback.print_exc()
                retry = True
            if self._mname:
                from fma_amd.controller import metrics as _m
                _m.queue_work_duration_seconds().labels(
                    self._mname).observe(time.monotonic() - t0)
            if isinstance(retry, (int, float)) and not isinstance(retry, bool) \
                    and retry > 0:
                self.queue.forget(item)  # a scheduled wait is not a failure
                self.queue.add_after(item, float(retry))
            elif retry:
                self.queue.add_rate_limited(item)
            else:
                self.queue.forget(item)
            self.queue.done(item)

    def stop(self) -> None:
        self.queue.shut_down()
        for th in self.threads:
            th.join(timeout=2)


class InitialSyncTracker:
    """Tracks when every item present at start has been processed once."""

    def __init__(self, on_done: Callable[[], None]):
        self._pending: set = set()
        self._started = False
        self._fired = False
        self._lock = threading.Lock()
        self._on_done = on_done

    def register(self, item) -> None:
        with self._lock:
            if not self._started:
                self._pending.add(item)

    def start(self) -> None:
        fire = False
        with self._lock:
            self._started = True
            fire = not self._pending and not self._fired
            if fire:
                self._fired = True
        if fire:
            self._on_done()

    def mark_processed(self, item) -> None:
        fire = False
        with self._lock:
            self._pending.discard(item)
            if self._started and not self._pending and not self._fired:
                self._fired = True
                fire = True
        if fire:
            self._on_done()


class _NodeItemState:
    __slots__ = ("add_time", "process_after")

    def __init__(self, add_time: float, process_after: float):
        self.add_time = add_time
        self.process_after = process_after


class TwoLevelQueue(Generic[T]):
    """Two-level per-node queue (reference pkg/controller/dual-pods/
    controller.go:404-424 nodeData, add/addAfter/takeReadyItems/
    earliestPending :1044-1097; nodeItem.process drains ready items
    oldest-first, inference-server.go:92-143).

    The outer rate-limited queue holds node names; each node keeps a local
    map of items with their add-time (for oldest-first drain) and
    process-after (for timed retries). The outer queue's dirty/processing
    sets guarantee a node is handled by at most one worker at a time, and
    per-item backoff is tracked per node — so a hot node retries on its own
    clock and cannot starve other nodes of worker time.
    """

    def __init__(self, node_of: Callable[[T], str],
                 max_backoff: float = MAX_BACKOFF_SECONDS,
                 metrics_name: str = "") -> None:
        self.node_of = node_of
        self.max_backoff = max_backoff
        self.outer: RateLimitingQueue[str] = RateLimitingQueue(
            max_backoff, metrics_name=metrics_name)
        self._mu = threading.Lock()
        self._nodes: dict = {}       # node -> {item: _NodeItemState}
        self._failures: dict = {}    # (node, item) -> consecutive failures

    def add(self, item: T, delay: float = 0.0) -> None:
        node = self.node_of(item)
        now = time.monotonic()
        with self._mu:
            d = self._nodes.setdefault(node, {})
            st = d.get(item)
            if st is None:
                d[item] = _NodeItemState(now, now + delay)
            else:
                # keep the original add time (oldest-first is by first
                # enqueue) but never push process-after later
                st.process_after = min(st.process_after, now + delay)
        if delay <= 0:
            self.outer.add(node)
        else:
            self.outer.add_after(node, delay)

    def add_after(self, item: T, delay: float) -> None:
        self.add(item, delay)

    def add_rate_limited(self, item: T) -> None:
        node = self.node_of(item)
        with self._mu:
            n = self._failures.get((node, item), 0)
            self._failures[(node, item)] = n + 1
        self.add(item, min(BASE_BACKOFF_SECONDS * (2 ** n),
                           self.max_backoff))

    def forget(self, item: T) -> None:
        node = self.node_of(item)
        with self._mu:
            self._failures.pop((node, item), None)

    def take_ready(self, node: str) -> List[T]:
        """Pop every item on the node whose process-after has passed,
        oldest first (reference takeReadyItems + the drain loop)."""
        now = time.monotonic()
        with self._mu:
            d = self._nodes.get(node)
            if not d:
                return []
            ready = [i for i, st in d.items() if st.process_after <= now]
            ready.sort(key=lambda i: d[i].add_time)
            for i in ready:
                d.pop(i)
            if not d:
                self._nodes.pop(node, None)
            return ready

    def earliest_pending(self, node: str) -> Optional[float]:
        """Seconds until the node's next not-yet-ready item (reference
        earliestPending), or None if the node has no pending items."""
        now = time.monotonic()
        with self._mu:
            d = self._nodes.get(node)
            if not d:
                return None
            return max(min(st.process_after for st in d.values()) - now, 0.0)

    def shut_down(self) -> None:
        self.outer.shut_down()

    def __len__(self) -> int:
        with self._mu:
            return sum(len(d) for d in self._nodes.values())


class NodeQueueAndWorkers(Generic[T]):
    """TwoLevelQueue + N workers; one worker drains one node at a time.

    `process(item)` keeps the flat-item signature (the node is embedded in
    the item; `node_of` extracts it) and the same return protocol as
    QueueAndWorkers: False done, True exponential retry, float = re-queue
    after exactly that many seconds."""

    def __init__(self, name: str, num_workers: int,
                 process: Callable[[T], bool],
                 node_of: Callable[[T], str],
                 max_backoff: float = MAX_BACKOFF_SECONDS,
                 metrics_name: str = ""):
        self.name = name
        self.queue: TwoLevelQueue[T] = TwoLevelQueue(
            node_of, max_backoff, metrics_name=metrics_name)
        self._mname = metrics_name
        self.num_workers = num_workers
        self.process = process
        self.threads: List[threading.Thread] = []

    def start(self) -> None:
        for i in range(self.num_workers):
            th = threading.Thread(target=self._worker, daemon=True,
                                  name=f"{self.name}-worker-{i}")
            th.start()
            self.threads.append(th)

    def _worker(self) -> None:
        while True:
            node = self.queue.outer.get()
            if node is None:
                return
            if self._mname:
                from fma_amd.controller import metrics as _m
                # depth = pending ITEMS across nodes (the fma_dpc_
                # innerqueue contract), not outer-queue nodes
                _m.queue_depth().labels(self._mname).set(len(self.queue))
            for item in self.queue.take_ready(node):
                t0 = time.monotonic()
                try:
                    retry = self.process(item)
                except Exception:  # noqa: BLE001 - keep the worker alive
                    import traceback
                    traceback.print_exc()
                    retry = True
                if self._mname:
                    from fma_amd.controller import metrics as _m
                    _m.queue_work_duration_seconds().labels(
                        self._mname).observe(time.monotonic() - t0)
                if isinstance(retry, (int, float)) \
                        and not isinstance(retry, bool) and retry > 0:
                    self.queue.forget(item)  # scheduled wait, not a failure
                    self.queue.add(item, float(retry))
                elif retry:
                    self.queue.add_rate_limited(item)
                else:
                    self.queue.forget(item)
            delay = self.queue.earliest_pending(node)
            self.queue.outer.done(node)
            if delay is not None:
                self.queue.outer.add_after(node, delay)

    def stop(self) -> None:
        self.queue.shut_down()
        for th in self.threads:
            th.join(timeout=2)
