"""GCS-hosted pub/sub topic bus (reference: src/ray/pubsub/publisher.h
GcsPublisher / subscriber.h GcsSubscriber — the channel bus carrying
log/error/node/actor events).

    from ray_amd.util import pubsub

    sub = pubsub.Subscriber("alerts")           # any process in the cluster
    pubsub.publish("alerts", {"sev": "high"})   # any other process
    msg = sub.poll(timeout=5)                   # -> {"sev": "high"}

Delivery is fan-out to every currently-subscribed process; there is no
replay for late subscribers (same semantics as the reference's GCS bus).
"""
from __future__ import annotations

import queue
from typing import Any, Optional

from .._core import runtime as _rtmod


def _rt():
    return _rtmod.global_runtime()


def publish(channel: str, data: Any) -> int:
    """Publish `data` (any msgpack-serializable value) to `channel`.
    Returns how many subscribers were reached."""
    return _rt().pubsub_publish(channel, data)


class Subscriber:
    """Queue-backed subscription to one channel."""

    def __init__(self, channel: str, maxsize: int = 10000):
        self.channel = channel
        self._q: "queue.Queue" = queue.Queue(maxsize)
        self._closed = False

        def _cb(data):
            try:
                self._q.put_nowait(data)
            except queue.Full:
                pass  # drop-oldest would need a lock; drop-newest is fine

        self._cb = _cb
        _rt().pubsub_subscribe(channel, _cb)

    def poll(self, timeout: Optional[float] = None) -> Any:
        """Block until the next message (raises queue.Empty on timeout)."""
        return self._q.get(timeout=timeout)

    def try_poll(self) -> Optional[Any]:
        try:
            return self._q.get_nowait()
        except queue.Empty:
            return None

    def close(self):
        if self._closed:
            return
        self._closed = True
        try:
            _rt().pubsub_unsubscribe(self.channel, self._cb)
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
