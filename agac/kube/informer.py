"""Shared informers and listers.

Replaces client-go's SharedInformerFactory (reference
``pkg/manager/manager.go:52-53`` uses two factories with a 30s resync, plus
the generated CRD informers under ``pkg/client/informers``).  Each informer
runs a list+watch loop against a ``KubeClient``, maintains a thread-safe
local cache, fans events out to registered handlers, and re-delivers
update(obj, obj) for every cached object each resync period (the
level-trigger that makes controllers self-healing).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, Optional, Tuple

from ..apis import meta as metalib
from .store import GoneError, NotFoundError

logger = logging.getLogger(__name__)


class Lister:
    """Namespace-scoped read access to an informer cache
    (client-go corelisters.ServiceLister etc.)."""

    def __init__(self, informer: "Informer", namespace: Optional[str] = None):
        self._informer = informer
        self._namespace = namespace

    def namespaced(self, namespace: str) -> "Lister":
        return Lister(self._informer, namespace)

    def get(self, name: str, namespace: Optional[str] = None):
        ns = namespace if namespace is not None else (self._namespace or "")
        obj = self._informer.cache_get(ns, name)
        if obj is None:
            raise NotFoundError(f"{self._informer.kind} {ns}/{name} not found in cache")
        return obj

    def list(self):
        return [
            o
            for o in self._informer.cache_list()
            if self._namespace is None or o.metadata.namespace == self._namespace
        ]


class Informer:
    """A shared informer for one kind."""

    def __init__(self, client, kind: str, resync_period: float = 30.0):
        self.client = client
        self.kind = kind
        self.resync_period = resync_period
        self._cache: Dict[Tuple[str, str], object] = {}
        self._cache_lock = threading.RLock()
        self._handlers = []
        self._synced = threading.Event()
        self._watch = None
        self._thread: Optional[threading.Thread] = None
        self._resync_thread: Optional[threading.Thread] = None
        self._stop: Optional[threading.Event] = None

    # -- registration ------------------------------------------------------
    def add_event_handler(self, on_add=None, on_update=None, on_delete=None):
        """Handlers: on_add(obj), on_update(old, new), on_delete(obj).

        READ-ONLY CONTRACT (same as client-go shared informers): handlers
        receive the informer's cached objects directly — they MUST NOT
        mutate them.  Controllers that need a mutable object go through a
        Lister Get, which returns a private deep copy."""
        self._handlers.append((on_add, on_update, on_delete))

    def lister(self) -> Lister:
        return Lister(self)

    def has_synced(self) -> bool:
        return self._synced.is_set()

    # -- cache -------------------------------------------------------------
    def cache_get(self, namespace: str, name: str):
        with self._cache_lock:
            obj = self._cache.get((namespace, name))
            return metalib.deep_copy(obj) if obj is not None else None

    def cache_list(self):
        with self._cache_lock:
            return [metalib.deep_copy(o) for o in self._cache.values()]

    # -- dispatch ----------------------------------------------------------
    def _dispatch_add(self, obj):
        for on_add, _, _ in self._handlers:
            if on_add:
                self._safe(on_add, obj)

    def _dispatch_update(self, old, new):
        for _, on_update, _ in self._handlers:
            if on_update:
                self._safe(on_update, old, new)

    def _dispatch_delete(self, obj):
        for _, _, on_delete in self._handlers:
            if on_delete:
                self._safe(on_delete, obj)

    @staticmethod
    def _safe(fn, *args):
        try:
            fn(*args)
        except Exception:  # utilruntime.HandleCrash: log, never kill the loop
            logger.exception("informer event handler panicked")

    # -- run loop ----------------------------------------------------------
    def run(self, stop: threading.Event):
        """Starts the list/watch loop and the resync timer. Idempotent."""
        if self._thread is not None:
            return
        self._stop = stop
        self._thread = threading.Thread(
            target=self._run_loop, name=f"informer-{self.kind}", daemon=True
        )
        self._thread.start()
        if self.resync_period and self.resync_period > 0:
            self._resync_thread = threading.Thread(
                target=self._resync_loop, name=f"informer-{self.kind}-resync", daemon=True
            )
            self._resync_thread.start()

    def stop(self):
        if self._watch is not None:
            try:
                self._watch.stop()
            except Exception:
                pass

    def _run_loop(self):
        stop = self._stop
        while not stop.is_set():
            try:
                self._list_and_watch(stop)
            except GoneError:
                # resourceVersion expired (apiserver 410) — relist quietly
                logger.info("informer %s watch expired; relisting", self.kind)
            except Exception:
                logger.exception("informer %s list/watch failed; backing off", self.kind)
                stop.wait(1.0)

    def _list_and_watch(self, stop: threading.Event):
        items, rv = self.client.list(self.kind)
        with self._cache_lock:
            fresh = {(o.metadata.namespace, o.metadata.name): o for o in items}
            old_cache = self._cache
            self._cache = fresh
        # Deliver deltas vs the previous cache contents (first run: all
        # adds).  Cached objects are handed to handlers directly — see the
        # read-only contract on add_event_handler.
        for key, obj in fresh.items():
            old = old_cache.get(key)
            if old is None:
                self._dispatch_add(obj)
            else:
                self._dispatch_update(old, obj)
        for key, obj in old_cache.items():
            if key not in fresh:
                self._dispatch_delete(obj)
        self._synced.set()

        # Watch loop: a closed stream re-watches from the last delivered
        # resourceVersion (real apiservers close watches every few minutes;
        # client-go resumes without relisting).  Only an expired rv
        # (GoneError, apiserver 410) escapes to the caller for a relist.
        last_rv = rv
        while not stop.is_set():
            watch = self.client.watch(self.kind, resource_version=last_rv)
            self._watch = watch
            try:
                while not stop.is_set():
                    event = watch.get(timeout=0.2)
                    if event is None:
                        if getattr(watch, "closed", False):
                            break  # stream ended → re-watch from last_rv
                        continue
                    self._handle_event(event)
                    if event.resource_version > last_rv:
                        last_rv = event.resource_version
            finally:
                watch.stop()
                self._watch = None

    def _handle_event(self, event):
        if event.type == "BOOKMARK":
            # metadata-only rv advancement (apiserver watch bookmarks) —
            # the caller already moved last_rv forward; nothing to cache
            # or dispatch (client-go reflector parity)
            return
        obj = event.obj
        key = (obj.metadata.namespace, obj.metadata.name)
        if event.type == "DELETED":
            with self._cache_lock:
                old = self._cache.pop(key, None)
            self._dispatch_delete(old if old is not None else obj)
        elif event.type in ("ADDED", "MODIFIED"):
            with self._cache_lock:
                old = self._cache.get(key)
                self._cache[key] = obj
            # handlers get the cached object itself (read-only contract)
            if old is None:
                self._dispatch_add(obj)
            else:
                self._dispatch_update(old, obj)

    def _resync_loop(self):
        stop = self._stop
        while not stop.wait(self.resync_period):
            with self._cache_lock:
                snapshot = list(self._cache.values())
            for obj in snapshot:
                # update(obj, obj) with the SAME cached object on both
                # sides, exactly like client-go resync — the DeepEqual /
                # rv-shortcut guard in the handlers drops these unless a
                # controller opts into cloud-resync re-enqueue
                self._dispatch_update(obj, obj)


class SharedInformerFactory:
    """One shared informer per kind (reference informers.SharedInformerFactory,
    ``pkg/manager/manager.go:52-53``)."""

    def __init__(self, client, resync_period: float = 30.0):
        self.client = client
        self.resync_period = resync_period
        self._informers: Dict[str, Informer] = {}
        self._lock = threading.Lock()
        self._started = False
        self._stop: Optional[threading.Event] = None

    def informer_for(self, kind: str) -> Informer:
        with self._lock:
            informer = self._informers.get(kind)
            if informer is None:
                informer = Informer(self.client, kind, self.resync_period)
                self._informers[kind] = informer
                if self._started:
                    informer.run(self._stop)
            return informer

    def services(self) -> Informer:
        return self.informer_for("Service")

    def ingresses(self) -> Informer:
        return self.informer_for("Ingress")

    def endpoint_group_bindings(self) -> Informer:
        return self.informer_for("EndpointGroupBinding")

    def start(self, stop: threading.Event):
        with self._lock:
            self._started = True
            self._stop = stop
            for informer in self._informers.values():
                informer.run(stop)


def wait_for_cache_sync(stop: threading.Event, *informers, timeout: float = 30.0) -> bool:
    """cache.WaitForCacheSync equivalent."""
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if stop.is_set():
            return False
        if all(i.has_synced() for i in informers):
            return True
        time.sleep(0.01)
    return all(i.has_synced() for i in informers)
