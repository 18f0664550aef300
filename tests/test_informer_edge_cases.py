"""Informer resilience: watch-log expiry → relist (410 Gone), watch failure
recovery, delta detection on relist, store field immutability."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.kube.client import InMemoryKubeClient
from agac.kube.informer import SharedInformerFactory, wait_for_cache_sync
from agac.kube.store import APIStore, GoneError


def mk_service(name, ns="default"):
    return corev1.Service(metadata=ObjectMeta(name=name, namespace=ns))


def wait_until(pred, timeout=10.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.01)
    return pred()


class TestWatchLogExpiry:
    def test_stale_rv_raises_gone_when_log_overflowed(self):
        store = APIStore()
        store.create(mk_service("first"))
        _, old_rv = store.list("Service")
        # overflow the bounded event log
        import agac.kube.store as store_mod

        for i in range(store_mod._EVENT_LOG_SIZE + 10):
            obj = store.get("Service", "default", "first")
            obj.metadata.annotations["i"] = str(i)
            store.update(obj)
        with pytest.raises(GoneError):
            store.watch("Service", resource_version=old_rv)

    def test_fresh_rv_ok_after_overflow(self):
        store = APIStore()
        store.create(mk_service("first"))
        import agac.kube.store as store_mod

        for i in range(store_mod._EVENT_LOG_SIZE + 10):
            obj = store.get("Service", "default", "first")
            obj.metadata.annotations["i"] = str(i)
            store.update(obj)
        _, rv = store.list("Service")
        watch = store.watch("Service", resource_version=rv)
        store.create(mk_service("second"))
        event = watch.get(timeout=10.0)
        assert event.obj.metadata.name == "second"
        watch.stop()


class TestRelistRecovery:
    def test_informer_survives_watch_failure_and_relists(self):
        """Kill the informer's live watch out from under it; the run loop
        must relist and deliver deltas accumulated in the gap."""
        client = InMemoryKubeClient()
        factory = SharedInformerFactory(client, resync_period=0)
        informer = factory.services()
        events = []
        informer.add_event_handler(
            on_add=lambda o: events.append(("add", o.metadata.name)),
            on_delete=lambda o: events.append(("del", o.metadata.name)),
        )
        stop = threading.Event()
        factory.start(stop)
        try:
            assert wait_for_cache_sync(stop, informer)
            client.create(mk_service("a"))
            assert wait_until(lambda: ("add", "a") in events)

            # sever the watch; mutate during the gap
            informer._watch.stop()
            client.create(mk_service("b"))
            client.delete("Service", "default", "a")

            # the relist must reconcile cache + emit delta events
            assert wait_until(lambda: ("add", "b") in events)
            assert wait_until(lambda: ("del", "a") in events)
            assert {o.metadata.name for o in informer.lister().list()} == {"b"}
        finally:
            stop.set()


class TestStoreFieldImmutability:
    def test_cannot_clear_deletion_timestamp_via_update(self):
        store = APIStore()
        binding_like = corev1.Service(
            metadata=ObjectMeta(name="s", namespace="default", finalizers=["f"])
        )
        store.create(binding_like)
        store.delete("Service", "default", "s")
        obj = store.get("Service", "default", "s")
        assert obj.metadata.deletion_timestamp is not None
        obj.metadata.deletion_timestamp = None  # attempt to resurrect
        store.update(obj)
        assert (
            store.get("Service", "default", "s").metadata.deletion_timestamp
            is not None
        )

    def test_uid_and_creation_timestamp_immutable(self):
        store = APIStore()
        store.create(mk_service("s"))
        original = store.get("Service", "default", "s")
        obj = store.get("Service", "default", "s")
        obj.metadata.uid = "spoofed"
        obj.metadata.creation_timestamp = "1999-01-01T00:00:00Z"
        store.update(obj)
        stored = store.get("Service", "default", "s")
        assert stored.metadata.uid == original.metadata.uid
        assert stored.metadata.creation_timestamp == original.metadata.creation_timestamp

    def test_namespaced_watch_filters(self):
        store = APIStore()
        _, rv = store.list("Service")
        watch = store.watch("Service", namespace="ns1", resource_version=rv)
        store.create(mk_service("other", ns="ns2"))
        store.create(mk_service("mine", ns="ns1"))
        event = watch.get(timeout=10.0)
        assert event.obj.metadata.name == "mine"
        watch.stop()


class TestRewatchWithoutRelist:
    def test_severed_watch_resumes_from_last_rv(self):
        """After a watch is cut, the informer re-watches from its last seen
        resourceVersion (no relist): gap events arrive via replay, and no
        duplicate ADDED is dispatched for objects already in cache."""
        client = InMemoryKubeClient()
        factory = SharedInformerFactory(client, resync_period=0)
        informer = factory.services()
        events = []
        informer.add_event_handler(
            on_add=lambda o: events.append(("add", o.metadata.name)),
            on_update=lambda old, new: events.append(("upd", new.metadata.name)),
            on_delete=lambda o: events.append(("del", o.metadata.name)),
        )
        stop = threading.Event()
        factory.start(stop)
        try:
            assert wait_for_cache_sync(stop, informer)
            client.create(mk_service("a"))
            assert wait_until(lambda: ("add", "a") in events)

            informer._watch.stop()  # sever
            client.create(mk_service("b"))
            obj = client.get("Service", "default", "a")
            obj.metadata.annotations["x"] = "y"
            client.update(obj)

            assert wait_until(lambda: ("add", "b") in events)
            assert wait_until(lambda: ("upd", "a") in events)
            # replay path: "a" must not be re-ADDED (cache still has it)
            assert events.count(("add", "a")) == 1
        finally:
            stop.set()
