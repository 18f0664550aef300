"""Typed Kubernetes clients.

``KubeClient`` is the seam the whole framework is written against (the
equivalent of ``kubernetes.Interface`` + the generated CRD clientset in the
reference, ``pkg/client/clientset/versioned``).  ``InMemoryKubeClient`` binds
it to ``APIStore``; ``agac.kube.rest.RestKubeClient`` binds the same interface
to a real HTTP API server.
"""

from __future__ import annotations

from typing import Optional

from ..apis import admissionregistration as admissionv1
from ..apis import core as corev1
from ..apis import endpointgroupbinding as egbv1alpha1
from .store import APIStore

_KINDS = {
    "Service": corev1.Service,
    "Ingress": corev1.Ingress,
    "Event": corev1.Event,
    "Lease": corev1.Lease,
    "EndpointGroupBinding": egbv1alpha1.EndpointGroupBinding,
    "ValidatingWebhookConfiguration": admissionv1.ValidatingWebhookConfiguration,
}


def class_for_kind(kind: str):
    return _KINDS[kind]


class _NamespacedResource:
    """Typed per-namespace resource client (mirrors the generated typed
    clients, e.g. ``OperatorV1alpha1().EndpointGroupBindings(ns)``)."""

    def __init__(self, client: "KubeClient", kind: str, namespace: str):
        self._client = client
        self._kind = kind
        self._namespace = namespace

    def create(self, obj):
        obj.metadata.namespace = obj.metadata.namespace or self._namespace
        return self._client.create(obj)

    def get(self, name: str):
        return self._client.get(self._kind, self._namespace, name)

    def list(self):
        items, _ = self._client.list(self._kind, self._namespace)
        return items

    def update(self, obj):
        return self._client.update(obj)

    def update_status(self, obj):
        return self._client.update_status(obj)

    def delete(self, name: str):
        return self._client.delete(self._kind, self._namespace, name)

    def patch(self, name: str, patch: dict, subresource=None):
        return self._client.patch(self._kind, self._namespace, name, patch, subresource)


class KubeClient:
    """Abstract client: CRUD + list/watch for every kind in the scheme."""

    # -- raw verbs (implemented by subclasses) ----------------------------
    def create(self, obj):
        raise NotImplementedError

    def get(self, kind: str, namespace: str, name: str):
        raise NotImplementedError

    def list(self, kind: str, namespace: Optional[str] = None):
        raise NotImplementedError

    def update(self, obj):
        raise NotImplementedError

    def update_status(self, obj):
        raise NotImplementedError

    def delete(self, kind: str, namespace: str, name: str):
        raise NotImplementedError

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource=None):
        """JSON merge patch; implemented by every backend."""
        raise NotImplementedError

    def watch(self, kind: str, namespace: Optional[str] = None, resource_version=None):
        raise NotImplementedError

    # -- typed accessors ---------------------------------------------------
    def services(self, namespace: str) -> _NamespacedResource:
        return _NamespacedResource(self, "Service", namespace)

    def ingresses(self, namespace: str) -> _NamespacedResource:
        return _NamespacedResource(self, "Ingress", namespace)

    def events(self, namespace: str) -> _NamespacedResource:
        return _NamespacedResource(self, "Event", namespace)

    def leases(self, namespace: str) -> _NamespacedResource:
        return _NamespacedResource(self, "Lease", namespace)

    def endpoint_group_bindings(self, namespace: str) -> _NamespacedResource:
        return _NamespacedResource(self, "EndpointGroupBinding", namespace)


class InMemoryKubeClient(KubeClient):
    """KubeClient bound directly to an in-process APIStore (the hermetic
    test / bench backend; the analogue of the generated fake clientset at
    reference ``pkg/client/clientset/versioned/fake``)."""

    def __init__(self, store: Optional[APIStore] = None):
        self.store = store or APIStore()

    def create(self, obj):
        return self.store.create(obj)

    def get(self, kind: str, namespace: str, name: str):
        return self.store.get(kind, namespace, name)

    def list(self, kind: str, namespace: Optional[str] = None):
        return self.store.list(kind, namespace)

    def update(self, obj):
        return self.store.update(obj)

    def update_status(self, obj):
        return self.store.update_status(obj)

    def delete(self, kind: str, namespace: str, name: str):
        return self.store.delete(kind, namespace, name)

    def patch(self, kind: str, namespace: str, name: str, patch: dict,
              subresource=None):
        return self.store.patch(kind, namespace, name, patch, subresource)

    def watch(self, kind: str, namespace: Optional[str] = None, resource_version=None):
        return self.store.watch(kind, namespace, resource_version)
