"""Reconcile engine tests (reference pkg/reconcile/reconcile.go semantics).

The reference has no unit tests for this engine (SURVEY.md §4 gap); these
tests pin down every branch of the Result/error handling table.
"""

from agac.errors import NoRetryError
from agac.kube.store import NotFoundError
from agac.kube.workqueue import ItemExponentialFailureRateLimiter, RateLimitingQueue
from agac.reconcile import Result, process_next_work_item


class Obj:
    def __init__(self, name):
        self.name = name

    def __deepcopy__(self, memo):  # engine deep-copies before processing
        return Obj(self.name)


def make_queue():
    return RateLimitingQueue(
        rate_limiter=ItemExponentialFailureRateLimiter(0.001, 0.01), name="t"
    )


def pump(q, key_to_obj, on_delete, on_update):
    q.add("default/x")
    return process_next_work_item(q, key_to_obj, on_delete, on_update)


def test_create_or_update_owns_the_looked_up_object():
    """The copy-before-process contract moved to key_to_obj: listers return
    a private deep copy per Get (informer.cache_get), so the engine passes
    the lookup result through unchanged instead of copying a second time
    (the reference copies in reconcile.go:52 because its lister returns the
    shared cache pointer)."""
    q = make_queue()
    calls = []
    original = Obj("x")
    pump(q, lambda k: original, lambda k: Result(), lambda o: calls.append(o) or Result())
    assert len(calls) == 1
    assert calls[0] is original  # no redundant second copy

    # ...and the real key_to_obj path (a Lister) DOES hand out private
    # copies, so mutations by a process func never leak into the cache
    from agac.apis import core as corev1
    from agac.apis.meta import ObjectMeta
    from agac.kube.client import InMemoryKubeClient
    from agac.kube.informer import Informer

    client = InMemoryKubeClient()
    client.create(corev1.Service(metadata=ObjectMeta(name="x", namespace="default")))
    informer = Informer(client, "Service", resync_period=0)
    import threading

    stop = threading.Event()
    informer.run(stop)
    try:
        from agac.kube.informer import wait_for_cache_sync

        assert wait_for_cache_sync(stop, informer)
        a = informer.lister().get("x", namespace="default")
        b = informer.lister().get("x", namespace="default")
        assert a is not b
        a.metadata.annotations["mutated"] = "true"
        assert "mutated" not in informer.lister().get("x", namespace="default").metadata.annotations
    finally:
        stop.set()
        informer.stop()


def test_not_found_routes_to_delete():
    q = make_queue()
    calls = []

    def key_to_obj(key):
        raise NotFoundError()

    pump(q, key_to_obj, lambda k: calls.append(k) or Result(), lambda o: Result())
    assert calls == ["default/x"]


def test_lookup_error_is_not_delete():
    q = make_queue()
    deleted = []

    def key_to_obj(key):
        raise RuntimeError("store broken")

    assert pump(q, key_to_obj, lambda k: deleted.append(k) or Result(), lambda o: Result())
    assert deleted == []


def test_error_requeues_rate_limited():
    q = make_queue()

    def boom(obj):
        raise RuntimeError("transient")

    pump(q, lambda k: Obj("x"), lambda k: Result(), boom)
    q.done("default/x")
    item, shutdown = q.get(timeout=10.0)
    assert item == "default/x" and not shutdown
    assert q.num_requeues("default/x") >= 1


def test_no_retry_error_forgets():
    q = make_queue()

    def boom(obj):
        raise NoRetryError("permanent")

    pump(q, lambda k: Obj("x"), lambda k: Result(), boom)
    q.done("default/x")
    item, _ = q.get(timeout=0.05)
    assert item is None  # not requeued


def test_requeue_after_uses_add_after_and_forgets():
    q = make_queue()
    results = iter([Result(requeue=True, requeue_after=0.02), Result()])
    seen = []

    def process(obj):
        seen.append(obj.name)
        return next(results)

    pump(q, lambda k: Obj("x"), lambda k: Result(), process)
    q.done("default/x")
    assert q.num_requeues("default/x") == 0  # forgotten before AddAfter
    item, _ = q.get(timeout=10.0)
    assert item == "default/x"


def test_requeue_flag_rate_limits():
    q = make_queue()
    pump(q, lambda k: Obj("x"), lambda k: Result(), lambda o: Result(requeue=True))
    q.done("default/x")
    item, _ = q.get(timeout=10.0)
    assert item == "default/x"
    assert q.num_requeues("default/x") >= 1


def test_success_forgets():
    q = make_queue()
    pump(q, lambda k: Obj("x"), lambda k: Result(), lambda o: Result())
    q.done("default/x")
    assert q.num_requeues("default/x") == 0
    item, _ = q.get(timeout=0.05)
    assert item is None


def test_shutdown_stops_worker():
    q = make_queue()
    q.shut_down()
    assert not process_next_work_item(q, lambda k: None, lambda k: Result(), lambda o: Result())


def test_non_string_key_is_forgotten():
    q = make_queue()
    q.add(42)
    assert process_next_work_item(q, lambda k: None, lambda k: Result(), lambda o: Result())
    q.done(42)
    item, _ = q.get(timeout=0.05)
    assert item is None
