"""Watch-event broadcaster with monotonic revisions.

Implements the kubernetes-watch-style protocol of the reference launcher
(reference inference_server/launcher/launcher.py:71-146 and protocol doc
docs/launcher.md:375-408): every instance state change gets a revision;
watchers follow from a cursor; a cursor older than the bounded buffer
raises RevisionTooOld, which the API maps to 410 Gone so clients re-LIST
and resume from the list revision.

Thread model: watchers run inside the asyncio event loop; publishers
(``append``) may run in FastAPI's sync-handler threadpool.  Waiter wakeups
are therefore published through ``loop.call_soon_threadsafe`` whenever the
caller is off-loop, and the waiter list is guarded by a mutex — a plain
``asyncio.Event.set()`` from a foreign thread is not guaranteed to wake a
parked loop.  A watcher whose cursor falls behind the bounded buffer
mid-stream (slow consumer vs. eviction) gets RevisionTooOld instead of a
silent gap, mirroring the reference's offset<0 check inside the watch
loop (reference launcher.py:124-146).
"""

from __future__ import annotations

import asyncio
import threading
from typing import Any, AsyncIterator, Dict, List, Optional


class RevisionTooOld(Exception):
    pass


class EventBroadcaster:
    BUFFER_LIMIT = 1000

    def __init__(self) -> None:
        self.revision = 0
        self._events: List[Dict[str, Any]] = []
        self._first_buffered = 1  # revision of the oldest buffered event
        self._waiters: List[asyncio.Event] = []
        self._mu = threading.Lock()
        self.loop: Optional[asyncio.AbstractEventLoop] = None

    def attach_loop(self, loop: asyncio.AbstractEventLoop) -> None:
        self.loop = loop

    def next_revision(self) -> int:
        self.revision += 1
        return self.revision

    def _wake_waiters(self) -> None:
        with self._mu:
            waiters = list(self._waiters)
        for w in waiters:
            w.set()

    def append(self, event_type: str, instance_id: str, revision: int,
               detail: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
        ev = {"type": event_type, "instance_id": instance_id,
              "revision": revision}
        if detail:
            ev.update(detail)
        self._events.append(ev)
        if len(self._events) > self.BUFFER_LIMIT:
            dropped = len(self._events) - self.BUFFER_LIMIT
            self._first_buffered += dropped
            del self._events[:dropped]
        loop = self.loop
        in_loop = False
        if loop is not None:
            try:
                in_loop = asyncio.get_running_loop() is loop
            except RuntimeError:
                in_loop = False
        if loop is not None and not in_loop and not loop.is_closed():
            loop.call_soon_threadsafe(self._wake_waiters)
        else:
            self._wake_waiters()
        return ev

    @property
    def oldest_buffered_revision(self) -> int:
        return self._events[0]["revision"] if self._events else self.revision + 1

    def check_since(self, since: int) -> None:
        """Raises RevisionTooOld when `since` predates the buffer (the
        client must re-LIST and resume from the list revision)."""
        if since and self._events and since + 1 < self.oldest_buffered_revision:
            raise RevisionTooOld(
                f"revision {since} too old; oldest is "
                f"{self.oldest_buffered_revision}")

    async def watch(self, since: int = 0) -> AsyncIterator[Dict[str, Any]]:
        """Yield events with revision > since, then block for new ones.

        Raises RevisionTooOld mid-stream if buffer eviction overtakes the
        cursor (events were lost); the caller terminates the stream so the
        client re-LISTs.
        """
        self.check_since(since)
        cursor = since
        while True:
            if cursor and self._events \
                    and cursor + 1 < self._events[0]["revision"]:
                raise RevisionTooOld(
                    f"watch cursor {cursor} overtaken by eviction; oldest "
                    f"buffered is {self._events[0]['revision']}")
            batch = [e for e in self._events if e["revision"] > cursor]
            if batch:
                for ev in batch:
                    cursor = max(cursor, ev["revision"])
                    yield ev
                continue  # events may have arrived while suspended at yield
            waiter = asyncio.Event()
            with self._mu:
                self._waiters.append(waiter)
            try:
                # close the compute-batch → register-waiter race: anything
                # appended in between would otherwise be missed until the
                # *next* append wakes us
                if not any(e["revision"] > cursor for e in self._events):
                    await waiter.wait()
            finally:
                with self._mu:
                    self._waiters.remove(waiter)
