"""In-memory Kubernetes API server core: typed object store with optimistic
concurrency, finalizer-aware deletion, generation tracking and watch streams.

This is the stand-in for the kube-apiserver that the reference talks to via
client-go.  All controller-visible semantics the reference relies on are
modeled:

- monotonically increasing ``resourceVersion`` per store, stamped on writes;
  updates with a stale resourceVersion fail with ``ConflictError``
  (needed for correct leader election and status-update races);
- ``metadata.generation`` increments when ``spec`` changes (not on status
  updates) — the EndpointGroupBinding reconciler gates on
  ``observedGeneration == generation`` (reference ``egb/reconcile.go:157``);
- deleting an object with finalizers only sets ``deletionTimestamp``; the
  object is removed once the last finalizer is dropped via update
  (reference relies on this for ``egb/reconcile.go:36-97``);
- list+watch: ``list`` returns a consistent snapshot plus the store
  resourceVersion, ``watch`` replays events after a given resourceVersion
  from a bounded log, then streams live events (informer contract).
"""

from __future__ import annotations

import itertools
import json
import queue
import threading
import time
import uuid
from collections import deque
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..apis import meta as metalib

ADDED = "ADDED"
MODIFIED = "MODIFIED"
DELETED = "DELETED"

# Bounded watch-event log; watches older than this are told to relist
# (equivalent of apiserver's "410 Gone" / resourceVersion too old).
_EVENT_LOG_SIZE = 4096


class APIError(Exception):
    def __init__(self, message: str, code: int):
        super().__init__(message)
        self.code = code


class NotFoundError(APIError):
    def __init__(self, message: str = "not found"):
        super().__init__(message, 404)


class AlreadyExistsError(APIError):
    def __init__(self, message: str = "already exists"):
        super().__init__(message, 409)


class ConflictError(APIError):
    def __init__(self, message: str = "resourceVersion conflict"):
        super().__init__(message, 409)


class GoneError(APIError):
    """resourceVersion too old — caller must relist."""

    def __init__(self, message: str = "resourceVersion too old"):
        super().__init__(message, 410)


def is_not_found(err: BaseException | None) -> bool:
    return isinstance(err, NotFoundError)


@dataclass
class WatchEvent:
    type: str
    obj: object
    resource_version: int


class _Watch:
    """A single watch subscription; iterated by informers."""

    def __init__(self, store: "APIStore", kind: str, namespace: Optional[str]):
        self._queue: "queue.Queue[Optional[WatchEvent]]" = queue.Queue()
        self._store = store
        self._kind = kind
        self._namespace = namespace
        self._stopped = False
        self.closed = False  # set once the end-of-stream sentinel is seen

    def _accepts(self, kind: str, namespace: str) -> bool:
        if kind != self._kind:
            return False
        return self._namespace is None or namespace == self._namespace

    def _push(self, event: WatchEvent):
        self._queue.put(event)

    def stop(self):
        if not self._stopped:
            self._stopped = True
            self._store._remove_watch(self)
            self._queue.put(None)

    def __iter__(self):
        return self

    def __next__(self) -> WatchEvent:
        item = self._queue.get()
        if item is None:
            raise StopIteration
        return item

    def get(self, timeout: Optional[float] = None) -> Optional[WatchEvent]:
        """None means timeout OR end-of-stream; check ``closed`` to tell
        them apart (a closed watch makes the informer relist)."""
        if self.closed:
            return None
        try:
            item = self._queue.get(timeout=timeout)
        except queue.Empty:
            return None
        if item is None:
            self.closed = True
        return item


class APIStore:
    """Thread-safe multi-kind object store with watch semantics."""

    def __init__(self):
        self._lock = threading.RLock()
        self._rv = itertools.count(1)
        self._last_rv = 0  # highest rv issued so far (peekable, for bookmarks)
        # kind -> {(namespace, name) -> obj}
        self._objects: Dict[str, Dict[Tuple[str, str], object]] = {}
        self._watches: List[_Watch] = []
        self._event_log: deque = deque(maxlen=_EVENT_LOG_SIZE)
        # registered admission webhooks (agac.kube.admission), consulted
        # before create/update/delete commits — the apiserver side of
        # ValidatingWebhookConfiguration.  VWC OBJECTS applied to the store
        # are consulted too (dynamic admission); service references resolve
        # through webhook_service_resolver (None = unresolvable, which the
        # failurePolicy then governs).
        self.admission_webhooks: List = []
        self.webhook_service_resolver = None

    def _admission_active(self) -> bool:
        return bool(self.admission_webhooks) or bool(
            self._objects.get("ValidatingWebhookConfiguration")
        )

    def _admit(self, kind: str, operation: str, old, new, subresource=None):
        """Run admission outside the store lock (webhook calls may do
        network I/O); commit-time rv checks still serialize writers.
        Consults both programmatically registered hooks and any
        ValidatingWebhookConfiguration OBJECTS in the store (dynamic
        admission, like a real apiserver — agac/kube/dynamicadmission.py)."""
        old_d = new_d = None
        if self.admission_webhooks:
            old_d = metalib.to_dict(old) if old is not None else None
            new_d = metalib.to_dict(new) if new is not None else None
            for hook in self.admission_webhooks:
                hook.admit(kind, operation, old_d, new_d)
        if self._bucket("ValidatingWebhookConfiguration"):
            from . import dynamicadmission

            if old_d is None and old is not None:
                old_d = metalib.to_dict(old)
            if new_d is None and new is not None:
                new_d = metalib.to_dict(new)
            dynamicadmission.admit(self, kind, operation, old_d, new_d,
                                   subresource=subresource)

    # -- helpers -----------------------------------------------------------
    def _bucket(self, kind: str) -> Dict[Tuple[str, str], object]:
        return self._objects.setdefault(kind, {})

    def _next_rv(self) -> int:
        self._last_rv = next(self._rv)
        return self._last_rv

    def latest_rv(self) -> int:
        """Highest resourceVersion issued so far, without burning one.
        Safe as a watch BOOKMARK rv: every event ≤ this rv has already been
        broadcast (rv issue and broadcast happen under the same lock)."""
        with self._lock:
            return self._last_rv

    def _broadcast(self, kind: str, event_type: str, obj, rv: int):
        event = WatchEvent(event_type, obj, rv)
        self._event_log.append((kind, event))
        for w in list(self._watches):
            if w._accepts(kind, obj.metadata.namespace):
                w._push(WatchEvent(event_type, metalib.deep_copy(obj), rv))

    def _remove_watch(self, w: _Watch):
        with self._lock:
            if w in self._watches:
                self._watches.remove(w)

    # -- CRUD --------------------------------------------------------------
    def create(self, obj):
        kind = type(obj).kind
        from .validation import validate_object

        validate_object(obj)
        self._admit(kind, "CREATE", None, obj)
        with self._lock:
            key = (obj.metadata.namespace, obj.metadata.name)
            bucket = self._bucket(kind)
            if key in bucket:
                raise AlreadyExistsError(f"{kind} {key} already exists")
            stored = metalib.deep_copy(obj)
            rv = self._next_rv()
            stored.metadata.resource_version = str(rv)
            stored.metadata.uid = stored.metadata.uid or str(uuid.uuid4())
            stored.metadata.creation_timestamp = (
                stored.metadata.creation_timestamp
                or time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
            )
            if hasattr(stored, "spec"):
                stored.metadata.generation = 1
            bucket[key] = stored
            self._broadcast(kind, ADDED, stored, rv)
            return metalib.deep_copy(stored)

    def get(self, kind: str, namespace: str, name: str):
        with self._lock:
            obj = self._bucket(kind).get((namespace, name))
            if obj is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            return metalib.deep_copy(obj)

    def list(self, kind: str, namespace: Optional[str] = None):
        """Returns (objects, resourceVersion) — a consistent snapshot."""
        with self._lock:
            items = [
                metalib.deep_copy(o)
                for (ns, _), o in sorted(self._bucket(kind).items())
                if namespace is None or ns == namespace
            ]
            # store-level rv: highest issued so far
            rv = self._next_rv()  # burn one to get a strictly usable marker
            return items, rv

    def list_page(
        self,
        kind: str,
        namespace: Optional[str] = None,
        limit: Optional[int] = None,
        continue_token: Optional[str] = None,
    ):
        """Chunked list, apiserver-style (``?limit=&continue=``).

        Returns (objects, resourceVersion, next_continue_token | None).
        Every page of one logical list carries the SAME resourceVersion
        (the one minted for the first page, carried in the opaque token).
        A continuation whose rv window has been evicted from the event log
        raises ``GoneError`` (the apiserver's 410 reason=Expired), telling
        the client to restart the list from scratch — exactly the
        client-go reflector contract."""
        import base64

        with self._lock:
            after: Tuple[str, str] = ("", "")
            if continue_token:
                try:
                    decoded = json.loads(
                        base64.urlsafe_b64decode(continue_token.encode()).decode()
                    )
                    rv = int(decoded["rv"])
                    after = (decoded["ns"], decoded["name"])
                except Exception:
                    raise GoneError("malformed continue token")
                # Expire continuations that predate the replayable window:
                # objects created/changed since the first page would be
                # invisibly skipped, and the client can no longer watch from
                # the page rv either.
                logged = [e for (k, e) in self._event_log if k == kind]
                if (
                    logged
                    and logged[0].resource_version > rv + 1
                    and len(self._event_log) == self._event_log.maxlen
                ):
                    raise GoneError("continue token expired")
            else:
                rv = self._next_rv()

            keys = sorted(
                key
                for key in self._bucket(kind)
                if namespace is None or key[0] == namespace
            )
            remaining = [k for k in keys if k > after]
            page_keys = remaining if limit is None else remaining[:limit]
            items = [metalib.deep_copy(self._bucket(kind)[k]) for k in page_keys]
            next_token = None
            if limit is not None and len(remaining) > limit:
                ns, name = page_keys[-1]
                next_token = base64.urlsafe_b64encode(
                    json.dumps({"rv": rv, "ns": ns, "name": name}).encode()
                ).decode()
            return items, rv, next_token

    def _check_rv(self, existing, obj):
        if (
            obj.metadata.resource_version
            and obj.metadata.resource_version != existing.metadata.resource_version
        ):
            raise ConflictError(
                f"{type(obj).kind} {obj.metadata.namespace}/{obj.metadata.name}: "
                f"resourceVersion {obj.metadata.resource_version} != "
                f"{existing.metadata.resource_version}"
            )

    def update(self, obj):
        """Update spec+metadata.  Bumps generation if spec changed; removes the
        object if it has a deletionTimestamp and finalizers became empty."""
        kind = type(obj).kind
        from .validation import validate_object

        validate_object(obj)
        if self._admission_active():
            try:
                current = self.get(kind, obj.metadata.namespace, obj.metadata.name)
            except NotFoundError:
                current = None
            self._admit(kind, "UPDATE", current, obj)
        with self._lock:
            key = (obj.metadata.namespace, obj.metadata.name)
            bucket = self._bucket(kind)
            existing = bucket.get(key)
            if existing is None:
                raise NotFoundError(f"{kind} {key} not found")
            self._check_rv(existing, obj)

            stored = metalib.deep_copy(obj)
            # immutable/system-owned fields
            stored.metadata.uid = existing.metadata.uid
            stored.metadata.creation_timestamp = existing.metadata.creation_timestamp
            stored.metadata.deletion_timestamp = existing.metadata.deletion_timestamp
            stored.metadata.generation = existing.metadata.generation
            if hasattr(stored, "status"):
                stored.status = metalib.deep_copy(existing.status)
            if hasattr(stored, "spec") and stored.spec != existing.spec:
                stored.metadata.generation += 1

            if (
                stored.metadata.deletion_timestamp is not None
                and not stored.metadata.finalizers
            ):
                rv = self._next_rv()
                stored.metadata.resource_version = str(rv)
                del bucket[key]
                self._broadcast(kind, DELETED, stored, rv)
                return metalib.deep_copy(stored)

            rv = self._next_rv()
            stored.metadata.resource_version = str(rv)
            bucket[key] = stored
            self._broadcast(kind, MODIFIED, stored, rv)
            return metalib.deep_copy(stored)

    def update_status(self, obj):
        """Status-subresource update: only ``status`` is taken from ``obj``.
        Admission applies only to webhooks whose rules name the status
        subresource ("<plural>/status"), like a real apiserver."""
        kind = type(obj).kind
        if self._admission_active():
            try:
                current = self.get(kind, obj.metadata.namespace, obj.metadata.name)
            except NotFoundError:
                current = None
            self._admit(kind, "UPDATE", current, obj, subresource="status")
        with self._lock:
            key = (obj.metadata.namespace, obj.metadata.name)
            bucket = self._bucket(kind)
            existing = bucket.get(key)
            if existing is None:
                raise NotFoundError(f"{kind} {key} not found")
            self._check_rv(existing, obj)
            stored = metalib.deep_copy(existing)
            stored.status = metalib.deep_copy(obj.status)
            rv = self._next_rv()
            stored.metadata.resource_version = str(rv)
            bucket[key] = stored
            self._broadcast(kind, MODIFIED, stored, rv)
            return metalib.deep_copy(stored)

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource: str | None = None):
        """JSON merge patch (RFC 7386) applied to the stored object; retries
        the optimistic-concurrency loop a few times on interleaved writes."""
        from .client import class_for_kind
        from .patch import json_merge_patch

        cls = class_for_kind(kind)
        for _ in range(5):
            existing = self.get(kind, namespace, name)
            merged_dict = json_merge_patch(metalib.to_dict(existing), patch)
            merged = metalib.from_dict(cls, merged_dict)
            merged.metadata.namespace = namespace
            merged.metadata.name = name
            merged.metadata.resource_version = existing.metadata.resource_version
            try:
                if subresource == "status":
                    return self.update_status(merged)
                return self.update(merged)
            except ConflictError:
                continue
        raise ConflictError(f"{kind} {namespace}/{name}: patch retries exhausted")

    def delete(self, kind: str, namespace: str, name: str):
        """Finalizer-aware delete (kube-apiserver graceful deletion)."""
        if self._admission_active():
            try:
                current = self.get(kind, namespace, name)
            except NotFoundError:
                current = None
            self._admit(kind, "DELETE", current, None)
        with self._lock:
            key = (namespace, name)
            bucket = self._bucket(kind)
            existing = bucket.get(key)
            if existing is None:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            if existing.metadata.finalizers:
                if existing.metadata.deletion_timestamp is None:
                    stored = metalib.deep_copy(existing)
                    stored.metadata.deletion_timestamp = time.strftime(
                        "%Y-%m-%dT%H:%M:%SZ", time.gmtime()
                    )
                    rv = self._next_rv()
                    stored.metadata.resource_version = str(rv)
                    bucket[key] = stored
                    self._broadcast(kind, MODIFIED, stored, rv)
                return None
            rv = self._next_rv()
            removed = metalib.deep_copy(existing)
            removed.metadata.resource_version = str(rv)
            del bucket[key]
            self._broadcast(kind, DELETED, removed, rv)
            return None

    # -- snapshot / restore -------------------------------------------------
    def dump(self) -> dict:
        """Serializable snapshot of every object + the rv counter (the
        apiserver's checkpoint; controllers themselves are stateless and
        resume from LIST, see docs/ARCHITECTURE.md)."""
        with self._lock:
            objects = [
                {"kind": kind, "object": metalib.to_dict(obj)}
                for kind, bucket in self._objects.items()
                for obj in bucket.values()
            ]
            # peek the counter without burning (itertools.count has no peek:
            # burn one and record it as the floor for the restored counter)
            next_rv = self._next_rv()
            return {"version": 1, "nextResourceVersion": next_rv, "objects": objects}

    @classmethod
    def load(cls, snapshot: dict) -> "APIStore":
        from .client import class_for_kind

        store = cls()
        next_rv = int(snapshot.get("nextResourceVersion", 1))
        store._rv = itertools.count(next_rv)
        # keep latest_rv() (bookmark source) monotonic across a restore
        store._last_rv = max(0, next_rv - 1)
        for entry in snapshot.get("objects", []):
            obj_cls = class_for_kind(entry["kind"])
            obj = metalib.from_dict(obj_cls, entry["object"])
            key = (obj.metadata.namespace, obj.metadata.name)
            store._bucket(entry["kind"])[key] = obj
        return store

    # -- checkpointing ------------------------------------------------------
    def save_snapshot(self, path: str):
        """Atomic snapshot write (tmp + rename)."""
        import json as jsonlib
        import os

        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            jsonlib.dump(self.dump(), f)
        os.replace(tmp, path)

    def start_checkpointer(self, path: str, interval: float, stop) -> "threading.Thread":
        """Periodic crash-resilient snapshots (the shutdown-only dump loses
        everything on a crash); returns the daemon thread.  A final
        snapshot on clean shutdown is still the caller's job."""
        def loop():
            while not stop.wait(interval):
                try:
                    self.save_snapshot(path)
                except Exception:  # never kill the server over a disk hiccup
                    import logging

                    logging.getLogger(__name__).exception(
                        "periodic snapshot to %s failed", path
                    )

        thread = threading.Thread(target=loop, name="store-checkpointer", daemon=True)
        thread.start()
        return thread

    # -- watch -------------------------------------------------------------
    def watch(
        self,
        kind: str,
        namespace: Optional[str] = None,
        resource_version: Optional[int] = None,
    ) -> _Watch:
        """Subscribe to events for ``kind``.  If ``resource_version`` is given,
        replays logged events with rv > resource_version first; raises
        ``GoneError`` if that window has been evicted from the log."""
        with self._lock:
            w = _Watch(self, kind, namespace)
            if resource_version is not None:
                logged = [
                    (k, e) for (k, e) in self._event_log if k == kind
                ]
                if logged and logged[0][1].resource_version > resource_version + 1:
                    # The requested window may predate the log.  Only an issue
                    # if events were actually evicted; be conservative.
                    if len(self._event_log) == self._event_log.maxlen:
                        raise GoneError()
                for _, e in logged:
                    if e.resource_version > resource_version and (
                        namespace is None or e.obj.metadata.namespace == namespace
                    ):
                        w._push(
                            WatchEvent(
                                e.type, metalib.deep_copy(e.obj), e.resource_version
                            )
                        )
            self._watches.append(w)
            return w
