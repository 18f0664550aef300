"""JSON merge patch (RFC 7386) across every backend: store semantics,
HTTP server routes (native + k8s), REST and k8s clients — the safe verb
for partial updates that must not clobber unmodeled fields."""

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.fixture import endpoint_group_binding
from agac.kube.client import InMemoryKubeClient
from agac.kube.httpapi import APIServer
from agac.kube.k8s import K8sKubeClient
from agac.kube.kubeconfig import RestConfig
from agac.kube.patch import json_merge_patch
from agac.kube.rest import RestKubeClient
from agac.kube.store import APIStore, NotFoundError


class TestRFC7386:
    def test_merge_recurses(self):
        target = {"a": {"x": 1, "y": 2}, "b": 3}
        patch = {"a": {"y": 9, "z": 8}}
        assert json_merge_patch(target, patch) == {"a": {"x": 1, "y": 9, "z": 8}, "b": 3}

    def test_none_deletes(self):
        assert json_merge_patch({"a": 1, "b": 2}, {"a": None}) == {"b": 2}

    def test_lists_replace(self):
        assert json_merge_patch({"a": [1, 2]}, {"a": [3]}) == {"a": [3]}

    def test_scalar_over_dict(self):
        assert json_merge_patch({"a": {"x": 1}}, {"a": 5}) == {"a": 5}

    def test_inputs_not_mutated(self):
        target = {"a": {"x": 1}}
        json_merge_patch(target, {"a": {"x": 2}})
        assert target == {"a": {"x": 1}}


class TestStorePatch:
    def test_patch_annotations_preserves_spec(self):
        store = APIStore()
        store.create(
            corev1.Service(
                metadata=ObjectMeta(name="s", namespace="d"),
                spec=corev1.ServiceSpec(
                    type="LoadBalancer",
                    ports=[corev1.ServicePort(port=80, protocol="TCP")],
                ),
            )
        )
        store.patch("Service", "d", "s", {"metadata": {"annotations": {"k": "v"}}})
        got = store.get("Service", "d", "s")
        assert got.metadata.annotations == {"k": "v"}
        assert got.spec.ports[0].port == 80
        assert got.metadata.generation == 1  # no spec change

    def test_patch_spec_bumps_generation(self):
        store = APIStore()
        store.create(endpoint_group_binding(name="b"))
        store.patch("EndpointGroupBinding", "default", "b", {"spec": {"weight": 9}})
        got = store.get("EndpointGroupBinding", "default", "b")
        assert got.spec.weight == 9
        assert got.spec.endpoint_group_arn  # untouched
        assert got.metadata.generation == 2

    def test_patch_status_subresource(self):
        store = APIStore()
        store.create(endpoint_group_binding(name="b"))
        store.patch(
            "EndpointGroupBinding", "default", "b",
            {"status": {"endpointIds": ["arn:lb"]}}, subresource="status",
        )
        got = store.get("EndpointGroupBinding", "default", "b")
        assert got.status.endpoint_ids == ["arn:lb"]
        assert got.metadata.generation == 1

    def test_patch_missing_object(self):
        store = APIStore()
        with pytest.raises(NotFoundError):
            store.patch("Service", "d", "ghost", {})


@pytest.fixture
def api():
    server = APIServer(APIStore())
    server.start()
    yield server
    server.shutdown()


class TestPatchOverTheWire:
    def seed(self, client):
        client.create(
            corev1.Service(
                metadata=ObjectMeta(name="s", namespace="d"),
                spec=corev1.ServiceSpec(
                    type="LoadBalancer",
                    ports=[corev1.ServicePort(port=80, protocol="TCP")],
                ),
            )
        )

    def test_rest_client_patch(self, api):
        client = RestKubeClient(api.url)
        self.seed(client)
        updated = client.patch("Service", "d", "s", {"metadata": {"labels": {"a": "1"}}})
        assert updated.metadata.labels == {"a": "1"}
        assert updated.spec.ports[0].port == 80

    def test_k8s_client_patch(self, api):
        client = K8sKubeClient(RestConfig(host=api.url))
        self.seed(client)
        updated = client.patch("Service", "d", "s", {"spec": {"type": "NodePort"}})
        assert updated.spec.type == "NodePort"
        assert updated.spec.ports[0].port == 80  # merge, not replace

    def test_typed_accessor_patch(self):
        client = InMemoryKubeClient()
        self.seed(client)
        client.services("d").patch("s", {"metadata": {"annotations": {"x": "y"}}})
        assert client.get("Service", "d", "s").metadata.annotations == {"x": "y"}


class TestMergePatchProperties:
    """RFC 7386 algebraic properties, fuzzed with hypothesis."""

    def test_idempotence_and_null_removal(self):
        from hypothesis import given, settings
        from hypothesis import strategies as st

        from agac.kube.patch import json_merge_patch

        scalars = st.one_of(st.none(), st.booleans(), st.integers(),
                            st.text(max_size=8))
        docs = st.recursive(
            scalars,
            lambda children: st.dictionaries(
                st.text(max_size=5), children, max_size=4
            ),
            max_leaves=12,
        )

        @given(doc=docs, patch=docs)
        @settings(max_examples=200, deadline=None)
        def check(doc, patch):
            once = json_merge_patch(doc, patch)
            twice = json_merge_patch(once, patch)
            assert once == twice  # idempotent (RFC 7386 §2)
            if isinstance(patch, dict) and isinstance(once, dict):
                for k, v in patch.items():
                    if v is None:
                        assert k not in once  # null removes
                    elif not isinstance(v, dict):
                        assert once.get(k) == v  # scalar replaces

        check()
