"""Watch-stream consistency: applying the event stream to an empty cache
must reconstruct the store's final state exactly (the invariant every
informer depends on), under randomized operation sequences including
finalizer-deferred deletions."""

import random

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.kube.store import APIStore


def apply_events(watch, cache, drain_timeout=0.5):
    while True:
        event = watch.get(timeout=drain_timeout)
        if event is None:
            return
        key = (event.obj.metadata.namespace, event.obj.metadata.name)
        if event.type == "DELETED":
            cache.pop(key, None)
        else:
            cache[key] = event.obj


def run_random_ops(seed, n_ops=300):
    rng = random.Random(seed)
    store = APIStore()
    _, rv0 = store.list("EndpointGroupBinding")
    watch = store.watch("EndpointGroupBinding", resource_version=rv0)

    names = [f"b-{i}" for i in range(8)]
    for _ in range(n_ops):
        name = rng.choice(names)
        op = rng.random()
        try:
            if op < 0.35:
                store.create(
                    egb.EndpointGroupBinding(
                        metadata=ObjectMeta(
                            name=name,
                            namespace="default",
                            finalizers=["f"] if rng.random() < 0.5 else [],
                        ),
                        spec=egb.EndpointGroupBindingSpec(
                            endpoint_group_arn=f"arn:{rng.randint(0, 5)}"
                        ),
                    )
                )
            elif op < 0.6:
                obj = store.get("EndpointGroupBinding", "default", name)
                obj.metadata.annotations["i"] = str(rng.randint(0, 9))
                store.update(obj)
            elif op < 0.75:
                obj = store.get("EndpointGroupBinding", "default", name)
                obj.status.endpoint_ids = [f"e-{rng.randint(0, 3)}"]
                store.update_status(obj)
            elif op < 0.9:
                store.delete("EndpointGroupBinding", "default", name)
            else:
                # finalizer removal (completes a deferred deletion if any)
                obj = store.get("EndpointGroupBinding", "default", name)
                obj.metadata.finalizers = []
                store.update(obj)
        except Exception:
            pass  # NotFound / AlreadyExists races are part of the sequence

    cache = {}
    apply_events(watch, cache)
    watch.stop()

    final = {
        (o.metadata.namespace, o.metadata.name): o
        for o in store.list("EndpointGroupBinding")[0]
    }
    return store, cache, final


def test_event_stream_reconstructs_state_seed_1():
    _, cache, final = run_random_ops(1)
    assert set(cache) == set(final)
    for key in final:
        assert cache[key].metadata.resource_version == final[key].metadata.resource_version
        assert cache[key].spec == final[key].spec
        assert cache[key].status == final[key].status


def test_event_stream_reconstructs_state_many_seeds():
    for seed in range(2, 12):
        _, cache, final = run_random_ops(seed, n_ops=150)
        assert set(cache) == set(final), f"seed {seed} diverged"
        for key in final:
            assert (
                cache[key].metadata.resource_version
                == final[key].metadata.resource_version
            ), f"seed {seed} stale object for {key}"


def test_event_rv_monotonic_per_stream():
    store = APIStore()
    _, rv0 = store.list("Service")
    watch = store.watch("Service", resource_version=rv0)
    for i in range(30):
        store.create(
            corev1.Service(metadata=ObjectMeta(name=f"s-{i}", namespace="d"))
        )
        if i % 3 == 0:
            store.delete("Service", "d", f"s-{i}")
    last = rv0
    while True:
        event = watch.get(timeout=0.5)
        if event is None:
            break
        assert event.resource_version > last
        last = event.resource_version
    watch.stop()
