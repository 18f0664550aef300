"""Dynamic admission: ValidatingWebhookConfiguration consumed as an API
OBJECT (the real cluster mechanism, and what the reference's kind e2e
exercises at e2e/e2e_test.go:77-103): apply config/webhook/manifests.yaml,
resolve the service reference to a live `agac webhook` TLS server with the
caBundle doing the trust, and the store enforces ARN immutability on
writes — no programmatic hook registration anywhere."""

import base64
import subprocess

import pytest

from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta
from agac.kube.admission import AdmissionDeniedError
from agac.kube.apply import apply_yaml
from agac.kube.client import InMemoryKubeClient
from agac.kube.store import APIStore
from agac.webhook.server import WebhookServer


@pytest.fixture(scope="module")
def tls_webhook(tmp_path_factory):
    d = tmp_path_factory.mktemp("dyncerts")
    cert, key = d / "tls.crt", d / "tls.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "1",
         "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True,
    )
    server = WebhookServer(port=0, tls_cert_file=str(cert), tls_key_file=str(key))
    server.start()
    yield server, cert.read_bytes()
    server.shutdown()


def binding(name="dyn", arn="arn:aws:globalaccelerator::1:x", weight=10):
    return egb.EndpointGroupBinding(
        metadata=ObjectMeta(name=name, namespace="default"),
        spec=egb.EndpointGroupBindingSpec(endpoint_group_arn=arn, weight=weight),
    )


@pytest.fixture
def store(tls_webhook):
    server, ca_pem = tls_webhook
    store = APIStore()
    client = InMemoryKubeClient(store)
    manifest = open("config/webhook/manifests.yaml").read()
    # inject the caBundle the way cert-manager would on a live cluster
    manifest = manifest.replace(
        "  clientConfig:",
        "  clientConfig:\n    caBundle: " + base64.b64encode(ca_pem).decode(),
    )
    results = apply_yaml(client, manifest)
    assert results[0][0] == "created"

    # service-reference resolution (clusters use <name>.<ns>.svc DNS)
    def resolver(service_ref):
        assert service_ref.name == "aws-global-accelerator-controller-webhook"
        return f"https://127.0.0.1:{server.port}"

    store.webhook_service_resolver = resolver
    return store


def test_arn_update_rejected_via_applied_manifest(store):
    store.create(binding())
    live = store.get("EndpointGroupBinding", "default", "dyn")
    live.spec.endpoint_group_arn = "arn:changed"
    with pytest.raises(AdmissionDeniedError, match="immutable"):
        store.update(live)


def test_weight_update_allowed(store):
    store.create(binding(name="dyn2"))
    live = store.get("EndpointGroupBinding", "default", "dyn2")
    live.spec.weight = 99
    updated = store.update(live)
    assert updated.spec.weight == 99


def test_create_passes_validator(store):
    # CREATE is registered (ADVICE r1 fix) and the validator allows it
    created = store.create(binding(name="dyn3"))
    assert created.metadata.name == "dyn3"


def test_unmatched_kind_skips_webhook(store):
    from agac.apis import core as corev1

    store.create(corev1.Service(metadata=ObjectMeta(name="svc", namespace="default")))


def test_unresolvable_service_honors_failure_policy(tls_webhook):
    server, ca_pem = tls_webhook
    store = APIStore()
    client = InMemoryKubeClient(store)
    manifest = open("config/webhook/manifests.yaml").read()
    apply_yaml(client, manifest)
    # no resolver installed → service unresolvable → failurePolicy: Fail
    with pytest.raises(AdmissionDeniedError, match="failurePolicy=Fail"):
        store.create(binding(name="fp"))
    # flip the config to Ignore via the API (like kubectl patch would)
    vwc = store.get("ValidatingWebhookConfiguration", "",
                    "aws-global-accelerator-controller-validating-webhook")
    vwc.webhooks[0].failure_policy = "Ignore"
    store.update(vwc)
    created = store.create(binding(name="fp"))
    assert created.metadata.name == "fp"


def test_deleting_the_configuration_disables_admission(store):
    store.delete("ValidatingWebhookConfiguration", "",
                 "aws-global-accelerator-controller-validating-webhook")
    store.create(binding(name="dyn4"))
    live = store.get("EndpointGroupBinding", "default", "dyn4")
    live.spec.endpoint_group_arn = "arn:changed-freely"
    store.update(live)  # no webhook left to veto


def test_vwc_over_the_wire_activates_admission():
    """Cluster-scoped VWC round-trips the k8s REST surface and admission
    applies to subsequent wire writes (no resolver installed here, so the
    Fail policy rejects — proving the config is live server-side)."""
    from agac.kube.httpapi import APIServer
    from agac.kube.k8s import K8sKubeClient
    from agac.kube.kubeconfig import RestConfig

    server = APIServer(APIStore())
    server.start()
    try:
        client = K8sKubeClient(RestConfig(host=server.url))
        results = apply_yaml(client, open("config/webhook/manifests.yaml").read())
        assert results[0][0] == "created"
        # cluster-scoped read back over the wire
        vwc = client.get("ValidatingWebhookConfiguration", "",
                         "aws-global-accelerator-controller-validating-webhook")
        assert vwc.webhooks[0].failure_policy == "Fail"
        # admission is active server-side for wire writes
        import pytest as _pytest

        with _pytest.raises(AdmissionDeniedError, match="failurePolicy=Fail"):
            client.create(binding(name="wire-fp"))
        # unmatched kinds unaffected
        from agac.apis import core as corev1

        client.create(corev1.Service(metadata=ObjectMeta(name="ok", namespace="default")))
    finally:
        server.shutdown()


def test_wildcard_rules_match():
    from agac.apis.admissionregistration import RuleWithOperations
    from agac.kube import k8swire
    from agac.kube.dynamicadmission import _rule_matches

    gvr = k8swire.gvr_for_kind("EndpointGroupBinding")
    assert _rule_matches(
        RuleWithOperations(api_groups=["*"], operations=["*"], resources=["*"]),
        gvr, "DELETE",
    )
    assert not _rule_matches(
        RuleWithOperations(api_groups=["*"], operations=["CREATE"], resources=["*"]),
        gvr, "DELETE",
    )
    assert not _rule_matches(
        RuleWithOperations(api_groups=["apps"], operations=["*"], resources=["*"]),
        gvr, "CREATE",
    )
    svc = k8swire.gvr_for_kind("Service")
    assert _rule_matches(
        RuleWithOperations(api_groups=[""], operations=["UPDATE"],
                           resources=["services"]),
        svc, "UPDATE",
    )


def test_webhook_outage_blocks_finalizer_removal_until_policy_relaxed():
    """The classic cluster incident, reproduced end to end: a Fail-closed
    webhook goes down; the EGB controller can then no longer remove its
    finalizer (spec updates are vetoed), so deletes wedge — until the
    operator flips failurePolicy to Ignore, at which point the drain
    completes."""
    import threading
    import time

    from agac.cloudprovider.aws.client import FakeCloudFactory
    from agac.cloudprovider.fake import FakeAWSBackend
    from agac.kube.store import NotFoundError
    from agac.manager import ControllerConfig, Manager

    store = APIStore()
    client = InMemoryKubeClient(store)
    backend = FakeAWSBackend()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    try:
        assert manager.wait_until_ready()
        # seed an endpoint group + binding while admission is permissive
        acc = backend.ga.create_accelerator("ext")
        from agac.cloudprovider.aws import types as t

        listener = backend.ga.create_listener(
            acc.accelerator_arn, [t.PortRange(80, 80)], "TCP")
        group = backend.ga.create_endpoint_group(listener.listener_arn, "us-east-1")
        lb = backend.elbv2.create_load_balancer("wh", region="us-east-1")
        from agac.apis import core as corev1

        client.create(corev1.Service(
            metadata=ObjectMeta(name="wh", namespace="default"),
            spec=corev1.ServiceSpec(type="LoadBalancer"),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
                )
            ),
        ))
        client.create(egb.EndpointGroupBinding(
            metadata=ObjectMeta(name="whb", namespace="default"),
            spec=egb.EndpointGroupBindingSpec(
                endpoint_group_arn=group.endpoint_group_arn,
                service_ref=egb.ServiceReference(name="wh"),
            ),
        ))
        deadline = time.monotonic() + 15
        while True:
            g = backend.ga.describe_endpoint_group(group.endpoint_group_arn)
            if any(d.endpoint_id == lb.load_balancer_arn
                   for d in g.endpoint_descriptions):
                break
            assert time.monotonic() < deadline
            time.sleep(0.02)

        # the webhook "goes down": a Fail-closed VWC with no reachable
        # backend appears (matches UPDATE on endpointgroupbindings)
        apply_yaml(client, open("config/webhook/manifests.yaml").read())

        client.delete("EndpointGroupBinding", "default", "whb")
        time.sleep(1.0)  # controller churns: finalizer removal is vetoed
        live = client.get("EndpointGroupBinding", "default", "whb")
        assert live.metadata.deletion_timestamp is not None
        assert live.metadata.finalizers, "finalizer vanished despite Fail-closed webhook"

        # operator escape hatch: relax the policy
        vwc = client.get("ValidatingWebhookConfiguration", "",
                         "aws-global-accelerator-controller-validating-webhook")
        vwc.webhooks[0].failure_policy = "Ignore"
        client.update(vwc)

        deadline = time.monotonic() + 15
        while True:
            try:
                client.get("EndpointGroupBinding", "default", "whb")
            except NotFoundError:
                break
            assert time.monotonic() < deadline, "drain never completed after relax"
            time.sleep(0.05)
    finally:
        stop.set()


def test_status_subresource_rules(store):
    """Status writes consult admission only when a rule names
    <plural>/status (real apiserver subresource matching): the default
    manifest (spec-only rules) never blocks status updates, and a
    status-scoped Fail-closed rule does."""
    store.create(binding(name="sub"))
    live = store.get("EndpointGroupBinding", "default", "sub")
    live.status.endpoint_ids = ["arn:lb"]
    store.update_status(live)  # default manifest: unaffected

    # add a status-scoped webhook with an unreachable backend (Fail)
    from agac.apis.admissionregistration import (
        RuleWithOperations,
        ValidatingWebhook,
        ValidatingWebhookConfiguration,
        WebhookClientConfig,
    )

    store.create(ValidatingWebhookConfiguration(
        metadata=ObjectMeta(name="status-guard"),
        webhooks=[ValidatingWebhook(
            name="status.example.com",
            client_config=WebhookClientConfig(url="https://127.0.0.1:1/x"),
            rules=[RuleWithOperations(
                api_groups=["operator.h3poteto.dev"],
                operations=["UPDATE"],
                resources=["endpointgroupbindings/status"],
            )],
            failure_policy="Fail",
            timeout_seconds=1,
        )],
    ))
    live = store.get("EndpointGroupBinding", "default", "sub")
    live.status.endpoint_ids = ["arn:lb2"]
    with pytest.raises(AdmissionDeniedError):
        store.update_status(live)
    # spec updates don't match the status-scoped rule (only the default
    # manifest's spec rule applies, and the weight change is allowed)
    live = store.get("EndpointGroupBinding", "default", "sub")
    live.spec.weight = 44
    store.update(live)
