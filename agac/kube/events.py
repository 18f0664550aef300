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
    def __init__(self, client, component: str):
        self._client = client
        self.component = component
        self._lock = threading.Lock()
        # (kind, ns, name, type, reason, message) -> Event name
        self._seen = {}

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
        key = (ref.kind, ref.namespace, ref.name, event_type, reason, message)
        with self._lock:
            existing_name = self._seen.get(key)
            try:
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
            except Exception:
                # Event recording must never break reconciliation.
                logger.exception("failed to record event %r", reason)

    def eventf(self, obj, event_type: str, reason: str, fmt: str, *args):
        self.event(obj, event_type, reason, fmt % args if args else fmt)
