"""ARN-hint optimization: steady-state reconciles skip the
O(#accelerators × ListTags) discovery scan, with verified fallback on any
hint miss — behavior must be identical to the full scan."""

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend

REGION = "us-east-1"


@pytest.fixture
def env():
    backend = FakeAWSBackend()
    cloud = FakeCloudFactory(backend)(REGION)
    return backend, cloud


def mk_service(name="web"):
    return corev1.Service(
        metadata=ObjectMeta(name=name, namespace="default"),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=80, protocol="TCP")],
        ),
    )


def seed(backend, cloud, n_noise=5):
    """One managed accelerator + n unrelated ones."""
    lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
    for i in range(n_noise):
        backend.ga.create_accelerator(f"noise-{i}")
    svc = mk_service()
    arn, created, _ = cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION
    )
    assert created
    return svc, lb, arn


def test_hint_skips_discovery_scan(env):
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    backend.ga.call_counts.clear()
    arn2, created, retry = cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=arn,
    )
    assert arn2 == arn and not created and retry == 0
    counts = backend.ga.call_counts
    assert counts.get("list_accelerators", 0) == 0  # no full scan
    assert counts.get("describe_accelerator", 0) == 1  # the hint fetch
    # ListTags: hint verification + the drift predicate
    assert counts.get("list_tags_for_resource", 0) <= 2


def test_without_hint_scans(env):
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    backend.ga.call_counts.clear()
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION
    )
    assert backend.ga.call_counts.get("list_accelerators", 0) >= 1


def test_stale_hint_falls_back_and_recreates(env):
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    # the accelerator disappears out-of-band
    cloud.cleanup_global_accelerator(arn)
    arn2, created, _ = cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=arn,  # stale
    )
    assert created and arn2 != arn
    accs, _ = backend.ga.list_accelerators()
    managed = [a for a in accs if a.name == "service-default-web"]
    assert len(managed) == 1  # no duplicate


def test_hint_pointing_at_foreign_accelerator_is_rejected(env):
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    foreign = backend.ga.create_accelerator("foreign")
    arn2, created, _ = cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=foreign.accelerator_arn,  # wrong tags → fallback
    )
    assert arn2 == arn and not created
    # foreign accelerator untouched
    assert backend.ga.describe_accelerator(foreign.accelerator_arn).name == "foreign"


def test_hint_drift_repair_still_works(env):
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    svc.spec.ports[0].port = 443
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=arn,
    )
    listener = cloud.get_listener(arn)
    assert [p.from_port for p in listener.port_ranges] == [443]


def test_controller_uses_and_invalidates_hints():
    """End-to-end: second reconcile of the same service skips the scan;
    deleting the service clears the hint before cleanup."""
    import threading
    import time

    from agac.manager import ControllerConfig, Manager
    from agac.kube.client import InMemoryKubeClient

    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    try:
        assert manager.wait_until_ready()
        lb = backend.elbv2.create_load_balancer("hinted", region=REGION)
        svc = mk_service("hinted")
        svc.metadata.annotations = {
            "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
        }
        svc.status = corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
            )
        )
        client.create(svc)
        deadline = time.monotonic() + 10
        while not backend.ga.list_accelerators()[0]:
            assert time.monotonic() < deadline
            time.sleep(0.02)

        backend.ga.call_counts.clear()
        stored = client.get("Service", "default", "hinted")
        stored.spec.ports[0].port = 8443
        client.update(stored)
        deadline = time.monotonic() + 10
        while True:
            accs, _ = backend.ga.list_accelerators()
            listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
            if [p.from_port for p in listeners[0].port_ranges] == [8443]:
                break
            assert time.monotonic() < deadline
            time.sleep(0.02)
        # the reconcile for the port change used the hint — no full scan
        # (list_accelerators calls here come only from the test itself)
        assert backend.ga.call_counts.get("list_tags_for_resource", 0) <= 2

        client.delete("Service", "default", "hinted")
        deadline = time.monotonic() + 10
        while backend.ga.list_accelerators()[0]:
            assert time.monotonic() < deadline
            time.sleep(0.02)
    finally:
        stop.set()


def test_hint_for_other_hostname_falls_back(env):
    """a resource with two LB hostnames has one owned
    accelerator per hostname; a hint pointing at the *other* hostname's
    accelerator must be rejected (target-hostname tag mismatch) so the
    reconcile falls back to the reference's full scan instead of retagging
    and repointing the wrong accelerator."""
    from agac.cloudprovider.aws import global_accelerator as ga_mod

    backend, cloud = env
    svc, lb_a, arn_a = seed(backend, cloud)
    # second accelerator owned by the same resource but targeting hostname B
    # (as the reference would leave behind for a 2-hostname service)
    lb_b = backend.elbv2.create_load_balancer("otherlb", region=REGION)
    import agac.cloudprovider.aws.types as t

    acc_b = backend.ga.create_accelerator(
        "service-default-web-b",
        tags=[
            t.Tag(ga_mod.GLOBAL_ACCELERATOR_MANAGED_TAG_KEY, "true"),
            t.Tag(
                ga_mod.GLOBAL_ACCELERATOR_OWNER_TAG_KEY,
                ga_mod.accelerator_owner_tag_value("service", "default", "web"),
            ),
            t.Tag(ga_mod.GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY, lb_b.dns_name),
            t.Tag(ga_mod.GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY, "c"),
        ],
    )
    backend.ga.call_counts.clear()
    # reconcile hostname A with a hint pointing at accelerator B
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb_a.dns_name), "c", "mylb", REGION,
        hint_arn=acc_b.accelerator_arn,
    )
    # hint must NOT short-circuit: the full scan ran
    assert backend.ga.call_counts.get("list_accelerators", 0) >= 1
    # accelerator A still targets hostname A (scan path behavior preserved)
    tags_a = {t_.key: t_.value for t_ in backend.ga.list_tags_for_resource(arn_a)}
    assert tags_a[ga_mod.GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY] == lb_a.dns_name


def test_hint_outcomes_are_counted(env):
    """hint hits/misses are observable via
    agac_hint_total{controller,outcome}."""
    prometheus_client = pytest.importorskip("prometheus_client")
    from agac import metrics

    def val(outcome):
        return metrics.HINT_TOTAL.labels(
            controller="globalaccelerator", outcome=outcome
        )._value.get()

    backend, cloud = env
    svc, lb, arn = seed(backend, cloud)
    before_hit, before_stale, before_err = val("hit"), val("stale"), val("error")
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=arn,
    )
    assert val("hit") == before_hit + 1
    foreign = backend.ga.create_accelerator("foreign")
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=foreign.accelerator_arn,
    )
    assert val("stale") == before_stale + 1
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn="arn:aws:globalaccelerator::1:accelerator/nonexistent",
    )
    assert val("error") == before_err + 1


def test_steady_state_reconcile_is_o1_aws_calls(env):
    """a steady-state (no-drift) hinted reconcile issues a
    constant number of AWS calls regardless of fleet size."""
    backend, cloud = env
    svc, lb, arn = seed(backend, cloud, n_noise=50)
    backend.ga.call_counts.clear()
    cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c", "mylb", REGION,
        hint_arn=arn,
    )
    total_ga_calls = sum(backend.ga.call_counts.values())
    # describe_accelerator + list_tags (verify) + listener/endpoint-group
    # reads for the drift predicates — constant, independent of the 50 noise
    # accelerators
    assert total_ga_calls <= 8, backend.ga.call_counts
