"""Kubernetes Event recording.

Replaces client-go's record.EventBroadcaster/EventRecorder (the reference
wires one per controller, e.g. ``pkg/controller/globalaccelerator/
controller.go:55-58``).  Events are logged and created as Event objects via
the kube client; repeated (object, type, reason, message) tuples aggregate
into a single Event with an incremented ``count``, like the real recorder.
"""

from __future__ import annotations

import logging
import threading
import time
import uuid

from ..apis import core as corev1
from ..apis.meta import ObjectMeta

logger = logging.getLogger(__name__)


class EventRecorder:
    """Async, like client-go: record.EventBroadcaster buffers events on a
    channel and a background goroutine writes them to the API (with
    count-aggregation), so recording never blocks a reconcile on API I/O.
    ``flush()`` drains the queue — call it in tests that assert on Event
    objects immediately after recording."""

    def __init__(self, client, component: str):
        import queue as _queue

        self._client = client
        self.component = component
        # (kind, ns, name, type, reason, message) -> Event name
        self._seen = {}
        self._queue: "_queue.Queue" = _queue.Queue(maxsize=1024)
        self._thread = threading.Thread(
            target=self._writer, name=f"event-recorder-{component}", daemon=True
        )
        self._thread.start()

    def event(self, obj, event_type: str, reason: str, message: str):
        ref = corev1.ObjectReference(
            kind=type(obj).kind,
            namespace=obj.metadata.namespace,
            name=obj.metadata.name,
            uid=obj.metadata.uid,
        )
        logger.info(
            'Event(%s/%s): type=%r reason=%r %s',
            ref.namespace, ref.name, event_type, reason, message,
        )
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        try:
            self._queue.put_nowait((ref, event_type, reason, message, now))
        except Exception:
            # full buffer: drop, like client-go's broadcaster under pressure
            logger.warning("event buffer full; dropping event %r", reason)

    def _writer(self):
        while True:
            item = self._queue.get()
            if item is None:  # stop() sentinel
                self._queue.task_done()
                return
            try:
                self._write(*item)
            except Exception:
                # Event recording must never break anything.
                logger.exception("failed to record event %r", item[2])
            finally:
                self._queue.task_done()

    def _write(self, ref, event_type: str, reason: str, message: str, now: str):
        key = (ref.kind, ref.namespace, ref.name, event_type, reason, message)
        existing_name = self._seen.get(key)
        if existing_name is not None:
            ev = self._client.get("Event", ref.namespace, existing_name)
            ev.count += 1
            ev.last_timestamp = now
            self._client.update(ev)
            return
        name = f"{ref.name}.{uuid.uuid4().hex[:10]}"
        ev = corev1.Event(
            metadata=ObjectMeta(name=name, namespace=ref.namespace or "default"),
            involved_object=ref,
            reason=reason,
            message=message,
            type=event_type,
            source=corev1.EventSource(component=self.component),
            count=1,
            first_timestamp=now,
            last_timestamp=now,
        )
        self._client.create(ev)
        self._seen[key] = name

    def flush(self, timeout: float = 5.0) -> bool:
        """Block until every queued event has been written (or timeout).
        Returns True if fully drained."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self._queue.unfinished_tasks == 0:
                return True
            time.sleep(0.005)
        return self._queue.unfinished_tasks == 0

    def stop(self):
        """Terminate the writer thread after draining queued events
        (controllers call this at shutdown so repeated manager lifecycles
        — e.g. leadership churn — don't accumulate threads)."""
        try:
            self._queue.put_nowait(None)
        except Exception:
            pass

    def eventf(self, obj, event_type: str, reason: str, fmt: str, *args):
        self.event(obj, event_type, reason, fmt % args if args else fmt)
