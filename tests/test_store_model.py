"""Bounded model-checking of the store's graceful-deletion state machine.

kube-apiserver deletion semantics the EndpointGroupBinding finalizer flow
depends on (reference egb/reconcile.go:18-110): deleting an object with
finalizers only stamps deletionTimestamp; an update that empties the
finalizer list while deleting removes the object; deletion is sticky (no
resurrection); generation bumps on spec change only.  Every call order up
to a bounded length runs against an executable spec model; visible state
(exists / deleting / finalizers / generation) must match after every op.
"""

import itertools

from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.kube.store import APIStore, NotFoundError

FIN = "operator.h3poteto.dev/endpointgroupbindings"


class ModelObject:
    def __init__(self):
        self.exists = False
        self.deleting = False
        self.finalizers = []
        self.generation = 0
        self.spec_weight = 0


class Model:
    """Spec model of apiserver graceful deletion."""

    def __init__(self):
        self.o = ModelObject()

    def create(self):
        if self.o.exists:
            return "exists-error"
        self.o = ModelObject()
        self.o.exists = True
        self.o.generation = 1
        return "ok"

    def delete(self):
        if not self.o.exists:
            return "notfound"
        if self.o.finalizers:
            self.o.deleting = True
            return "ok"
        self.o.exists = False
        return "ok"

    def update(self, finalizers, weight):
        if not self.o.exists:
            return "notfound"
        if weight != self.o.spec_weight:
            self.o.generation += 1
            self.o.spec_weight = weight
        self.o.finalizers = list(finalizers)
        if self.o.deleting and not self.o.finalizers:
            self.o.exists = False
        return "ok"


def visible(store):
    try:
        obj = store.get("EndpointGroupBinding", "d", "x")
    except NotFoundError:
        return None
    return (
        obj.metadata.deletion_timestamp is not None,
        tuple(obj.metadata.finalizers),
        obj.metadata.generation,
    )


def model_visible(model):
    if not model.o.exists:
        return None
    return (model.o.deleting, tuple(model.o.finalizers), model.o.generation)


OPS = ["create", "delete", "upd_fin_w1", "upd_nofin_w1", "upd_fin_w2"]


def apply_op(store, model, op):
    if op == "create":
        try:
            store.create(egb.EndpointGroupBinding(
                metadata=ObjectMeta(name="x", namespace="d"),
                spec=egb.EndpointGroupBindingSpec(
                    endpoint_group_arn="arn:aws:globalaccelerator::1:x",
                    weight=1,
                ),
            ))
            real = "ok"
        except Exception:
            real = "exists-error"
        spec = model.create()
        if spec == "ok":
            model.o.spec_weight = 1
        assert real == spec, (op, real, spec)
    elif op == "delete":
        try:
            store.delete("EndpointGroupBinding", "d", "x")
            real = "ok"
        except NotFoundError:
            real = "notfound"
        assert real == model.delete(), op
    else:
        finalizers = [FIN] if "fin" in op.split("_")[1] and op != "upd_nofin_w1" else []
        weight = 2 if op.endswith("w2") else 1
        try:
            live = store.get("EndpointGroupBinding", "d", "x")
            live.metadata.finalizers = finalizers
            live.spec.weight = weight
            store.update(live)
            real = "ok"
        except NotFoundError:
            real = "notfound"
        assert real == model.update(finalizers, weight), op


def test_all_deletion_schedules_match_the_model():
    n = 0
    for length in range(1, 7):
        for schedule in itertools.product(OPS, repeat=length):
            store = APIStore()
            model = Model()
            for op in schedule:
                apply_op(store, model, op)
                assert visible(store) == model_visible(model), (
                    schedule, op, visible(store), model_visible(model)
                )
            n += 1
    assert n == sum(5**k for k in range(1, 7))  # 19,530 schedules


def test_deletion_is_sticky():
    """Once deleting, re-adding a finalizer cannot resurrect: the object
    still vanishes when finalizers empty, and deletionTimestamp persists."""
    store = APIStore()
    store.create(egb.EndpointGroupBinding(
        metadata=ObjectMeta(name="x", namespace="d", finalizers=[FIN]),
        spec=egb.EndpointGroupBindingSpec(
            endpoint_group_arn="arn:aws:globalaccelerator::1:x"),
    ))
    store.delete("EndpointGroupBinding", "d", "x")
    obj = store.get("EndpointGroupBinding", "d", "x")
    assert obj.metadata.deletion_timestamp is not None
    # finalizer churn while deleting
    obj.metadata.finalizers = [FIN, "other/finalizer"]
    obj = store.update(obj)
    assert obj.metadata.deletion_timestamp is not None  # sticky
    obj.metadata.finalizers = []
    store.update(obj)
    try:
        store.get("EndpointGroupBinding", "d", "x")
        raise AssertionError("object survived finalizer removal while deleting")
    except NotFoundError:
        pass
