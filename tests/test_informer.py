"""Informer / lister tests: list+watch, handler dispatch, cache sync, resync."""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.kube.client import InMemoryKubeClient
from agac.kube.informer import SharedInformerFactory, wait_for_cache_sync
from agac.kube.store import NotFoundError


def mk_service(name, ns="default"):
    return corev1.Service(
        metadata=ObjectMeta(name=name, namespace=ns),
        spec=corev1.ServiceSpec(type="LoadBalancer"),
    )


def wait_until(pred, timeout=5.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.01)
    return pred()


@pytest.fixture
def env():
    client = InMemoryKubeClient()
    factory = SharedInformerFactory(client, resync_period=0)
    stop = threading.Event()
    yield client, factory, stop
    stop.set()


def test_initial_list_dispatches_adds_and_syncs(env):
    client, factory, stop = env
    client.create(mk_service("a"))
    client.create(mk_service("b"))
    adds = []
    informer = factory.services()
    informer.add_event_handler(on_add=lambda o: adds.append(o.metadata.name))
    factory.start(stop)
    assert wait_for_cache_sync(stop, informer)
    assert wait_until(lambda: sorted(adds) == ["a", "b"])


def test_watch_add_update_delete(env):
    client, factory, stop = env
    events = []
    informer = factory.services()
    informer.add_event_handler(
        on_add=lambda o: events.append(("add", o.metadata.name)),
        on_update=lambda old, new: events.append(("update", new.metadata.name)),
        on_delete=lambda o: events.append(("del", o.metadata.name)),
    )
    factory.start(stop)
    assert wait_for_cache_sync(stop, informer)

    client.create(mk_service("x"))
    assert wait_until(lambda: ("add", "x") in events)

    obj = client.get("Service", "default", "x")
    obj.metadata.annotations["k"] = "v"
    client.update(obj)
    assert wait_until(lambda: ("update", "x") in events)

    client.delete("Service", "default", "x")
    assert wait_until(lambda: ("del", "x") in events)


def test_lister_reads_cache(env):
    client, factory, stop = env
    client.create(mk_service("a", ns="ns1"))
    client.create(mk_service("b", ns="ns2"))
    informer = factory.services()
    factory.start(stop)
    assert wait_for_cache_sync(stop, informer)
    lister = informer.lister()
    assert {o.metadata.name for o in lister.list()} == {"a", "b"}
    ns1 = lister.namespaced("ns1")
    assert [o.metadata.name for o in ns1.list()] == ["a"]
    assert ns1.get("a").metadata.namespace == "ns1"
    with pytest.raises(NotFoundError):
        ns1.get("b")


def test_lister_returns_copies(env):
    client, factory, stop = env
    client.create(mk_service("a"))
    informer = factory.services()
    factory.start(stop)
    assert wait_for_cache_sync(stop, informer)
    obj = informer.lister().namespaced("default").get("a")
    obj.metadata.annotations["mutated"] = "yes"
    again = informer.lister().namespaced("default").get("a")
    assert "mutated" not in again.metadata.annotations


def test_resync_redelivers_updates():
    client = InMemoryKubeClient()
    factory = SharedInformerFactory(client, resync_period=0.05)
    stop = threading.Event()
    try:
        client.create(mk_service("a"))
        updates = []
        informer = factory.services()
        informer.add_event_handler(
            on_update=lambda old, new: updates.append(new.metadata.name)
        )
        factory.start(stop)
        assert wait_for_cache_sync(stop, informer)
        assert wait_until(lambda: updates.count("a") >= 2)
    finally:
        stop.set()


def test_handler_exception_does_not_kill_informer(env):
    client, factory, stop = env
    good = []
    informer = factory.services()
    informer.add_event_handler(on_add=lambda o: 1 / 0)
    informer.add_event_handler(on_add=lambda o: good.append(o.metadata.name))
    factory.start(stop)
    assert wait_for_cache_sync(stop, informer)
    client.create(mk_service("a"))
    client.create(mk_service("b"))
    assert wait_until(lambda: sorted(good) == ["a", "b"])


def test_shared_informer_is_shared(env):
    _, factory, _ = env
    assert factory.services() is factory.services()
    assert factory.services() is not factory.ingresses()
