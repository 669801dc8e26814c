"""Watch-event broadcaster with monotonic revisions.

Implements the kubernetes-watch-style protocol of the reference launcher
(reference inference_server/launcher/launcher.py:71-146 and protocol doc
docs/launcher.md:375-408): every instance state change gets a revision;
watchers follow from a cursor; a cursor older than the bounded buffer
raises RevisionTooOld, which the API maps to 410 Gone so clients re-LIST
and resume from the list revision.
"""

from __future__ import annotations

import asyncio
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

    def next_revision(self) -> int:
        self.revision += 1
        return self.revision

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
        for w in self._waiters:
            w.set()
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
        """Yield events with revision > since, then block for new ones."""
        self.check_since(since)
        cursor = since
        while True:
            batch = [e for e in self._events if e["revision"] > cursor]
            for ev in batch:
                cursor = max(cursor, ev["revision"])
                yield ev
            waiter = asyncio.Event()
            self._waiters.append(waiter)
            try:
                await waiter.wait()
            finally:
                self._waiters.remove(waiter)
