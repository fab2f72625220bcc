"""In-process typed pub/sub (reference: controlplane/pubsub Topic[T] —
stateless pipe with non-blocking Publish, per-subscriber bounded buffer
with drop-oldest overflow, and panic-recovered delivery so a bad
subscriber can never take down PID 1). Python redesign: a Topic fans an
event out to per-subscription deques; readers pull with a timeout. Used
by cpd to push live events to `controlplane events -f` streams without
the publisher ever blocking on a slow client."""
from __future__ import annotations

import threading
from collections import deque
from typing import Any, Callable

from ..logger import get as get_logger

log = get_logger("pubsub")

DEFAULT_BUFFER = 256


class Subscription:
    """One subscriber's bounded buffer. get() blocks up to timeout;
    dropped counts events lost to overflow (drop-oldest)."""

    def __init__(self, topic: "Topic", buffer: int):
        self._topic = topic
        self._buf: deque = deque(maxlen=buffer)
        self._cond = threading.Condition()
        self.dropped = 0
        self.closed = False

    def _push(self, event: Any) -> None:
        with self._cond:
            if self.closed:
                return
            if len(self._buf) == self._buf.maxlen:
                self.dropped += 1
            self._buf.append(event)
            self._cond.notify()

    def get(self, timeout: float | None = None) -> Any | None:
        """Next event, or None on timeout/close."""
        with self._cond:
            if not self._buf:
                self._cond.wait(timeout)
            if self._buf:
                return self._buf.popleft()
            return None

    def close(self) -> None:
        with self._cond:
            self.closed = True
            self._cond.notify_all()
        self._topic._unsubscribe(self)


class Topic:
    def __init__(self, name: str, buffer: int = DEFAULT_BUFFER):
        self.name = name
        self._buffer = buffer
        self._subs: list[Subscription] = []
        self._lock = threading.Lock()

    def subscribe(self, buffer: int | None = None) -> Subscription:
        sub = Subscription(self, buffer or self._buffer)
        with self._lock:
            self._subs.append(sub)
        return sub

    def _unsubscribe(self, sub: Subscription) -> None:
        with self._lock:
            try:
                self._subs.remove(sub)
            except ValueError:
                pass

    def publish(self, event: Any) -> None:
        """Non-blocking: every live subscription gets the event (or drops
        its oldest); a failing subscriber is isolated."""
        with self._lock:
            subs = list(self._subs)
        for sub in subs:
            try:
                sub._push(event)
            except Exception as e:      # never propagate to the publisher
                log.error("pubsub_deliver_failed", topic=self.name, err=str(e))

    @property
    def subscriber_count(self) -> int:
        with self._lock:
            return len(self._subs)
