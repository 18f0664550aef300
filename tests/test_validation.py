"""CRD-schema validation at the store boundary + batched weight sync."""

import pytest

from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.fixture import endpoint_group_binding
from agac.kube.client import InMemoryKubeClient
from agac.kube.validation import ValidationError, validate_object


class TestEGBSchemaValidation:
    def test_valid_fixture_passes(self):
        validate_object(endpoint_group_binding())

    def test_missing_arn_rejected(self):
        binding = endpoint_group_binding(endpoint_group_arn="")
        with pytest.raises(ValidationError, match="endpointGroupArn"):
            validate_object(binding)

    def test_bad_weight_rejected(self):
        binding = endpoint_group_binding(weight=-1)
        with pytest.raises(ValidationError, match="int32"):
            validate_object(binding)
        binding = endpoint_group_binding()
        binding.spec.weight = 2**31
        with pytest.raises(ValidationError):
            validate_object(binding)
        binding.spec.weight = "heavy"
        with pytest.raises(ValidationError, match="integer"):
            validate_object(binding)

    def test_empty_ref_name_rejected(self):
        binding = endpoint_group_binding()
        binding.spec.service_ref = egb.ServiceReference(name="")
        with pytest.raises(ValidationError, match="serviceRef"):
            validate_object(binding)

    def test_store_rejects_invalid_create(self):
        client = InMemoryKubeClient()
        with pytest.raises(ValidationError):
            client.create(endpoint_group_binding(endpoint_group_arn=""))

    def test_store_rejects_invalid_update(self):
        client = InMemoryKubeClient()
        client.create(endpoint_group_binding(name="b"))
        stored = client.get("EndpointGroupBinding", "default", "b")
        stored.spec.weight = -5
        with pytest.raises(ValidationError):
            client.update(stored)

    def test_missing_name_rejected(self):
        client = InMemoryKubeClient()
        with pytest.raises(ValidationError, match="metadata.name"):
            client.create(
                egb.EndpointGroupBinding(
                    metadata=ObjectMeta(namespace="default"),
                    spec=egb.EndpointGroupBindingSpec(endpoint_group_arn="arn:x"),
                )
            )

    def test_invalid_over_k8s_wire_is_422_invalid(self):
        from agac.kube.httpapi import APIServer
        from agac.kube.k8s import K8sKubeClient
        from agac.kube.kubeconfig import RestConfig
        from agac.kube.store import APIStore

        api = APIServer(APIStore())
        api.start()
        try:
            client = K8sKubeClient(RestConfig(host=api.url))
            with pytest.raises(ValidationError):
                client.create(endpoint_group_binding(endpoint_group_arn=""))
        finally:
            api.shutdown()


class TestBatchedWeightSync:
    def test_sync_endpoint_weights_single_roundtrip(self):
        backend = FakeAWSBackend()
        cloud = FakeCloudFactory(backend)("us-east-1")
        acc = backend.ga.create_accelerator("a")
        listener = backend.ga.create_listener(acc.accelerator_arn, [t.PortRange(80, 80)], "TCP")
        group = backend.ga.create_endpoint_group(
            listener.listener_arn,
            "us-east-1",
            endpoint_configurations=[
                t.EndpointConfiguration(endpoint_id="arn:lb1", weight=1),
                t.EndpointConfiguration(endpoint_id="arn:lb2", weight=2),
                t.EndpointConfiguration(endpoint_id="arn:other", weight=3),
            ],
        )
        cloud.sync_endpoint_weights(group, ["arn:lb1", "arn:lb2"], 50)
        desc = backend.ga.describe_endpoint_group(group.endpoint_group_arn)
        weights = {d.endpoint_id: d.weight for d in desc.endpoint_descriptions}
        assert weights == {"arn:lb1": 50, "arn:lb2": 50, "arn:other": 3}


class TestAdmissionFailurePolicy:
    """ValidatingWebhookConfiguration failurePolicy parity: an unreachable
    webhook rejects writes under Fail (the k8s default) and admits under
    Ignore."""

    def _store_with_hook(self, failure_policy):
        from agac.apis import endpointgroupbinding as egb
        from agac.kube.admission import http_admission
        from agac.kube.store import APIStore

        store = APIStore()
        store.admission_webhooks.append(http_admission(
            kinds=["EndpointGroupBinding"],
            operations=["CREATE", "UPDATE"],
            url="http://127.0.0.1:1",  # nothing listens here
            timeout=0.2,
            failure_policy=failure_policy,
        ))
        from agac.apis.meta import ObjectMeta

        binding = egb.EndpointGroupBinding(
            metadata=ObjectMeta(name="fp", namespace="d"),
            spec=egb.EndpointGroupBindingSpec(
                endpoint_group_arn="arn:aws:globalaccelerator::1:x"),
        )
        return store, binding

    def test_fail_policy_rejects_when_webhook_down(self):
        import pytest

        from agac.kube.admission import AdmissionDeniedError

        store, binding = self._store_with_hook("Fail")
        with pytest.raises(AdmissionDeniedError, match="failurePolicy=Fail"):
            store.create(binding)

    def test_ignore_policy_admits_when_webhook_down(self):
        store, binding = self._store_with_hook("Ignore")
        created = store.create(binding)
        assert created.metadata.name == "fp"
