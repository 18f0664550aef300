"""In-memory API store tests: CRUD, optimistic concurrency, generation
tracking, finalizer-aware deletion, list+watch."""

import pytest

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta, from_dict, to_dict
from agac.kube.store import (
    APIStore,
    AlreadyExistsError,
    ConflictError,
    NotFoundError,
)


def mk_service(name="web", ns="default", **meta):
    return corev1.Service(
        metadata=ObjectMeta(name=name, namespace=ns, **meta),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=80, protocol="TCP")],
        ),
    )


def test_create_get_roundtrip():
    s = APIStore()
    created = s.create(mk_service())
    assert created.metadata.resource_version
    assert created.metadata.uid
    assert created.metadata.generation == 1
    got = s.get("Service", "default", "web")
    assert got.spec.ports[0].port == 80


def test_create_duplicate():
    s = APIStore()
    s.create(mk_service())
    with pytest.raises(AlreadyExistsError):
        s.create(mk_service())


def test_get_not_found():
    s = APIStore()
    with pytest.raises(NotFoundError):
        s.get("Service", "default", "nope")


def test_update_bumps_generation_on_spec_change_only():
    s = APIStore()
    s.create(mk_service())
    obj = s.get("Service", "default", "web")
    obj.metadata.annotations["x"] = "1"
    updated = s.update(obj)
    assert updated.metadata.generation == 1  # annotation only

    obj = s.get("Service", "default", "web")
    obj.spec.ports[0].port = 443
    updated = s.update(obj)
    assert updated.metadata.generation == 2


def test_update_conflict_on_stale_rv():
    s = APIStore()
    s.create(mk_service())
    a = s.get("Service", "default", "web")
    b = s.get("Service", "default", "web")
    a.metadata.annotations["a"] = "1"
    s.update(a)
    b.metadata.annotations["b"] = "2"
    with pytest.raises(ConflictError):
        s.update(b)


def test_update_does_not_touch_status():
    s = APIStore()
    s.create(mk_service())
    obj = s.get("Service", "default", "web")
    obj.status.load_balancer.ingress.append(corev1.LoadBalancerIngress(hostname="h"))
    s.update(obj)
    assert s.get("Service", "default", "web").status.load_balancer.ingress == []


def test_update_status_subresource():
    s = APIStore()
    s.create(mk_service())
    obj = s.get("Service", "default", "web")
    obj.status.load_balancer.ingress.append(corev1.LoadBalancerIngress(hostname="h"))
    s.update_status(obj)
    stored = s.get("Service", "default", "web")
    assert stored.status.load_balancer.ingress[0].hostname == "h"
    assert stored.metadata.generation == 1  # status never bumps generation


def test_delete_without_finalizers_removes():
    s = APIStore()
    s.create(mk_service())
    s.delete("Service", "default", "web")
    with pytest.raises(NotFoundError):
        s.get("Service", "default", "web")


def test_delete_with_finalizers_sets_deletion_timestamp():
    s = APIStore()
    binding = egb.EndpointGroupBinding(
        metadata=ObjectMeta(name="b", namespace="default", finalizers=["f"]),
        spec=egb.EndpointGroupBindingSpec(endpoint_group_arn="arn:x"),
    )
    s.create(binding)
    s.delete("EndpointGroupBinding", "default", "b")
    obj = s.get("EndpointGroupBinding", "default", "b")
    assert obj.metadata.deletion_timestamp is not None
    # dropping the finalizer via update removes the object
    obj.metadata.finalizers = []
    s.update(obj)
    with pytest.raises(NotFoundError):
        s.get("EndpointGroupBinding", "default", "b")


def test_list_and_watch_replay():
    s = APIStore()
    s.create(mk_service("a"))
    items, rv = s.list("Service")
    assert [o.metadata.name for o in items] == ["a"]
    w = s.watch("Service", resource_version=rv)
    s.create(mk_service("b"))
    ev = w.get(timeout=10.0)
    assert ev.type == "ADDED" and ev.obj.metadata.name == "b"
    w.stop()


def test_watch_sees_modify_and_delete():
    s = APIStore()
    s.create(mk_service())
    _, rv = s.list("Service")
    w = s.watch("Service", resource_version=rv)
    obj = s.get("Service", "default", "web")
    obj.metadata.annotations["k"] = "v"
    s.update(obj)
    ev = w.get(timeout=10.0)
    assert ev.type == "MODIFIED"
    s.delete("Service", "default", "web")
    ev = w.get(timeout=10.0)
    assert ev.type == "DELETED"
    w.stop()


def test_serialization_roundtrip_wire_format():
    svc = mk_service()
    d = to_dict(svc)
    assert d["spec"]["type"] == "LoadBalancer"
    assert d["spec"]["ports"][0]["port"] == 80
    back = from_dict(corev1.Service, d)
    assert back.spec.ports[0].port == 80


def test_egb_wire_format_field_names():
    binding = egb.EndpointGroupBinding(
        metadata=ObjectMeta(name="b", namespace="default"),
        spec=egb.EndpointGroupBindingSpec(
            endpoint_group_arn="arn:aws:globalaccelerator::123:accelerator/a/listener/l/endpoint-group/e",
            client_ip_preservation=True,
            weight=100,
            service_ref=egb.ServiceReference(name="web"),
        ),
        status=egb.EndpointGroupBindingStatus(endpoint_ids=["arn:lb"], observed_generation=2),
    )
    d = to_dict(binding)
    # CRD wire format parity (reference types.go json tags)
    assert d["spec"]["endpointGroupArn"].startswith("arn:aws")
    assert d["spec"]["clientIPPreservation"] is True
    assert d["spec"]["weight"] == 100
    assert d["spec"]["serviceRef"] == {"name": "web"}
    assert d["status"]["endpointIds"] == ["arn:lb"]
    assert d["status"]["observedGeneration"] == 2
    back = from_dict(egb.EndpointGroupBinding, d)
    assert back.spec.client_ip_preservation is True
    assert back.spec.weight == 100
