"""Global Accelerator resource-manager flows against the stateful AWS fake:
ensure (create/update/drift-repair), retry paths, cleanup, endpoint
management.  The reference cannot test any of this hermetically (no AWS
fake); this suite is the mocked-backend matrix BASELINE.json
demands."""

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.aws.errors import (
    EndpointGroupNotFoundException,
    LoadBalancerNotFoundException,
)
from agac.cloudprovider.fake import FakeAWSBackend

REGION = "us-east-1"


@pytest.fixture
def backend():
    return FakeAWSBackend()


@pytest.fixture
def cloud(backend):
    return FakeCloudFactory(backend)(REGION)


def mk_service(name="web", ns="default", ports=((80, "TCP"),), annotations=None):
    return corev1.Service(
        metadata=ObjectMeta(name=name, namespace=ns, annotations=dict(annotations or {})),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=p, protocol=proto) for p, proto in ports],
        ),
    )


def seed_lb(backend, name="mylb", state=t.LB_STATE_ACTIVE):
    return backend.elbv2.create_load_balancer(name, region=REGION, state=state)


def lb_ingress(lb):
    return corev1.LoadBalancerIngress(hostname=lb.dns_name)


class TestEnsureForService:
    def test_creates_triple_with_ownership_tags(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service()
        arn, created, retry = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "cluster1", "mylb", REGION
        )
        assert created and retry == 0
        acc = backend.ga.describe_accelerator(arn)
        assert acc.name == "service-default-web"
        assert acc.enabled
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(arn)}
        assert tags == {
            "aws-global-accelerator-controller-managed": "true",
            "aws-global-accelerator-owner": "service/default/web",
            "aws-global-accelerator-target-hostname": lb.dns_name,
            "aws-global-accelerator-cluster": "cluster1",
        }
        listener = cloud.get_listener(arn)
        assert [(p.from_port, p.to_port) for p in listener.port_ranges] == [(80, 80)]
        assert listener.protocol == "TCP"
        group = cloud.get_endpoint_group(listener.listener_arn)
        assert group.endpoint_group_region == REGION
        assert [d.endpoint_id for d in group.endpoint_descriptions] == [
            lb.load_balancer_arn
        ]

    def test_ensure_is_idempotent(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service()
        arn1, created1, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        arn2, created2, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert created1 and not created2
        assert arn1 == arn2
        accs, _ = backend.ga.list_accelerators()
        assert len(accs) == 1

    def test_lb_not_active_requeues_30s(self, backend, cloud):
        lb = seed_lb(backend, state=t.LB_STATE_PROVISIONING)
        svc = mk_service()
        arn, created, retry = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert arn is None and not created and retry == 30.0
        accs, _ = backend.ga.list_accelerators()
        assert accs == []
        # becomes active → creates
        backend.elbv2.set_state("mylb", t.LB_STATE_ACTIVE)
        arn, created, retry = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert created and retry == 0

    def test_dns_mismatch_errors(self, backend, cloud):
        seed_lb(backend)
        svc = mk_service()
        with pytest.raises(ValueError, match="DNS name is not matched"):
            cloud.ensure_global_accelerator_for_service(
                svc,
                corev1.LoadBalancerIngress(hostname="other.elb.us-east-1.amazonaws.com"),
                "c",
                "mylb",
                REGION,
            )

    def test_missing_lb_raises_typed_error(self, backend, cloud):
        svc = mk_service()
        with pytest.raises(LoadBalancerNotFoundException):
            cloud.ensure_global_accelerator_for_service(
                svc,
                corev1.LoadBalancerIngress(hostname="x.elb.us-east-1.amazonaws.com"),
                "c",
                "ghost",
                REGION,
            )

    def test_ip_address_type_annotation(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service(
            annotations={
                "aws-global-accelerator-controller.h3poteto.dev/ip-address-type": "ipv4"
            }
        )
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert backend.ga.describe_accelerator(arn).ip_address_type == "IPV4"

    def test_client_ip_preservation_annotation(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service(
            annotations={
                "aws-global-accelerator-controller.h3poteto.dev/client-ip-preservation": "true"
            }
        )
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        listener = cloud.get_listener(arn)
        group = cloud.get_endpoint_group(listener.listener_arn)
        assert group.endpoint_descriptions[0].client_ip_preservation_enabled is True


class TestUpdateDriftRepair:
    def ensure(self, backend, cloud, svc):
        lb = seed_lb(backend)
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        return lb, arn

    def test_port_drift_updates_listener(self, backend, cloud):
        svc = mk_service()
        lb, arn = self.ensure(backend, cloud, svc)
        svc.spec.ports.append(corev1.ServicePort(port=443, protocol="TCP"))
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        listener = cloud.get_listener(arn)
        assert sorted(p.from_port for p in listener.port_ranges) == [80, 443]

    def test_deleted_listener_is_recreated(self, backend, cloud):
        svc = mk_service()
        lb, arn = self.ensure(backend, cloud, svc)
        listener = cloud.get_listener(arn)
        group = cloud.get_endpoint_group(listener.listener_arn)
        backend.ga.delete_endpoint_group(group.endpoint_group_arn)
        backend.ga.delete_listener(listener.listener_arn)
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        listener = cloud.get_listener(arn)
        group = cloud.get_endpoint_group(listener.listener_arn)
        assert group.endpoint_descriptions[0].endpoint_id == lb.load_balancer_arn

    def test_disabled_accelerator_is_reenabled(self, backend, cloud):
        svc = mk_service()
        lb, arn = self.ensure(backend, cloud, svc)
        backend.ga.update_accelerator(arn, enabled=False)
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert backend.ga.describe_accelerator(arn).enabled

    def test_renamed_accelerator_is_renamed_back(self, backend, cloud):
        svc = mk_service()
        lb, arn = self.ensure(backend, cloud, svc)
        backend.ga.update_accelerator(arn, name="tampered")
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert backend.ga.describe_accelerator(arn).name == "service-default-web"

    def test_endpoint_group_membership_repaired(self, backend, cloud):
        svc = mk_service()
        lb, arn = self.ensure(backend, cloud, svc)
        listener = cloud.get_listener(arn)
        group = cloud.get_endpoint_group(listener.listener_arn)
        backend.ga.update_endpoint_group(
            group.endpoint_group_arn,
            endpoint_configurations=[t.EndpointConfiguration(endpoint_id="arn:other")],
        )
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        group = cloud.get_endpoint_group(listener.listener_arn)
        assert [d.endpoint_id for d in group.endpoint_descriptions] == [
            lb.load_balancer_arn
        ]


class TestCleanup:
    def test_cleanup_deletes_triple(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service()
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        cloud.cleanup_global_accelerator(arn)
        accs, _ = backend.ga.list_accelerators()
        assert accs == []

    def test_cleanup_missing_accelerator_is_noop(self, cloud):
        cloud.cleanup_global_accelerator(
            "arn:aws:globalaccelerator::123456789012:accelerator/ghost"
        )

    def test_cleanup_partial_triple(self, backend, cloud):
        # accelerator + listener but no endpoint group
        acc = backend.ga.create_accelerator("orphan")
        backend.ga.create_listener(
            acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
        )
        cloud.cleanup_global_accelerator(acc.accelerator_arn)
        accs, _ = backend.ga.list_accelerators()
        assert accs == []


class TestListByTags:
    def test_list_by_resource_and_hostname(self, backend, cloud):
        lb = seed_lb(backend)
        svc = mk_service()
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c1", "mylb", REGION
        )
        # unmanaged accelerator is ignored
        backend.ga.create_accelerator("manual")
        by_res = cloud.list_global_accelerator_by_resource("c1", "service", "default", "web")
        assert [a.accelerator_arn for a in by_res] == [arn]
        assert cloud.list_global_accelerator_by_resource("c2", "service", "default", "web") == []
        by_host = cloud.list_global_accelerator_by_hostname(lb.dns_name, "c1")
        assert [a.accelerator_arn for a in by_host] == [arn]
        assert cloud.list_global_accelerator_by_hostname("nope", "c1") == []

    def test_pagination(self, backend, cloud):
        for i in range(7):
            backend.ga.create_accelerator(f"acc-{i}")
        page, token = backend.ga.list_accelerators(max_results=3)
        assert len(page) == 3 and token is not None
        total = cloud._list_accelerators()
        assert len(total) == 7


class TestCreatePartialFailureCleanup:
    def test_listener_create_failure_rolls_back(self, backend, cloud, monkeypatch):
        lb = seed_lb(backend)
        svc = mk_service()

        def boom(*a, **k):
            raise RuntimeError("listener create failed")

        monkeypatch.setattr(backend.ga, "create_listener", boom)
        with pytest.raises(RuntimeError):
            cloud.ensure_global_accelerator_for_service(
                svc, lb_ingress(lb), "c", "mylb", REGION
            )
        # partial accelerator was cleaned up
        accs, _ = backend.ga.list_accelerators()
        assert accs == []


class TestEndpointManagement:
    def make_group(self, backend, cloud):
        acc = backend.ga.create_accelerator("a")
        listener = backend.ga.create_listener(
            acc.accelerator_arn, [t.PortRange(80, 80)], "TCP"
        )
        return backend.ga.create_endpoint_group(listener.listener_arn, REGION)

    def test_add_and_remove_lb(self, backend, cloud):
        group = self.make_group(backend, cloud)
        lb = seed_lb(backend, "lb1")
        endpoint_id, retry = cloud.add_lb_to_endpoint_group(group, "lb1", False, 128)
        assert retry == 0 and endpoint_id == lb.load_balancer_arn
        desc = cloud.describe_endpoint_group(group.endpoint_group_arn)
        assert desc.endpoint_descriptions[0].weight == 128
        cloud.remove_lb_from_endpoint_group(group, endpoint_id)
        desc = cloud.describe_endpoint_group(group.endpoint_group_arn)
        assert desc.endpoint_descriptions == []

    def test_add_lb_not_active_requeues(self, backend, cloud):
        group = self.make_group(backend, cloud)
        seed_lb(backend, "lb1", state=t.LB_STATE_PROVISIONING)
        endpoint_id, retry = cloud.add_lb_to_endpoint_group(group, "lb1", False, None)
        assert endpoint_id is None and retry == 30.0

    def test_update_weight_preserves_other_endpoints(self, backend, cloud):
        group = self.make_group(backend, cloud)
        lb1 = seed_lb(backend, "lb1")
        lb2 = seed_lb(backend, "lb2")
        cloud.add_lb_to_endpoint_group(group, "lb1", False, 10)
        cloud.add_lb_to_endpoint_group(group, "lb2", False, 20)
        cloud.update_endpoint_weight(group, lb1.load_balancer_arn, 99)
        desc = cloud.describe_endpoint_group(group.endpoint_group_arn)
        weights = {d.endpoint_id: d.weight for d in desc.endpoint_descriptions}
        # the fix over the reference: lb2 survives the weight update
        assert weights == {lb1.load_balancer_arn: 99, lb2.load_balancer_arn: 20}

    def test_describe_missing_group_raises_typed(self, cloud):
        with pytest.raises(EndpointGroupNotFoundException):
            cloud.describe_endpoint_group("arn:ghost")


class TestDeleteAcceleratorLifecycle:
    def test_disable_poll_delete(self, backend, cloud):
        # deploy_after_describes=1 → first describe IN_PROGRESS, then DEPLOYED;
        # the manager's poll loop must survive that.
        acc = backend.ga.create_accelerator("a")
        cloud._delete_accelerator(acc.accelerator_arn)
        accs, _ = backend.ga.list_accelerators()
        assert accs == []

    def test_delete_enabled_accelerator_rejected_by_fake(self, backend):
        acc = backend.ga.create_accelerator("a")
        from agac.cloudprovider.aws.errors import AcceleratorNotDisabledException

        with pytest.raises(AcceleratorNotDisabledException):
            backend.ga.delete_accelerator(acc.accelerator_arn)


class TestUserTagDrift:
    def test_tags_annotation_change_retags_accelerator(self, backend, cloud):
        svc = mk_service()
        lb = seed_lb(backend)
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        # user adds the tags annotation later → accelerator must be retagged
        svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-tags"
        ] = "env=prod,team=net"
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(arn)}
        assert tags["env"] == "prod"
        assert tags["team"] == "net"
        # ownership tags survive the retag
        assert tags["aws-global-accelerator-owner"] == "service/default/web"

    def test_name_annotation_change_renames(self, backend, cloud):
        svc = mk_service()
        lb = seed_lb(backend)
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-name"
        ] = "renamed"
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        assert backend.ga.describe_accelerator(arn).name == "renamed"

    def test_removed_user_tag_persists_like_reference(self, backend, cloud):
        """Shared quirk: TagResource merges and the drift predicate only
        checks target ⊆ actual (reference global_accelerator.go:426-436,
        :730-738), so a user tag removed from the annotation stays on the
        accelerator in both implementations."""
        svc = mk_service(
            annotations={
                "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-tags": "env=prod"
            }
        )
        lb = seed_lb(backend)
        arn, _, _ = cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        del svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-tags"
        ]
        cloud.ensure_global_accelerator_for_service(
            svc, lb_ingress(lb), "c", "mylb", REGION
        )
        tags = {x.key: x.value for x in backend.ga.list_tags_for_resource(arn)}
        assert tags.get("env") == "prod"  # stale but reference-faithful


def test_multi_hostname_last_wins_parity():
    """PARITY quirk 8b: two LB hostnames on one service converge to ONE
    accelerator tagged with the LAST hostname processed (each ensure's
    owner-scan update loop retags whatever the previous hostname's pass
    ensured — reference global_accelerator.go:130-141)."""
    from agac.apis import core as corev1
    from agac.apis.meta import ObjectMeta
    from agac.cloudprovider.aws.client import FakeCloudFactory
    from agac.cloudprovider.fake import FakeAWSBackend

    backend = FakeAWSBackend()
    cloud = FakeCloudFactory(backend)("us-east-1")
    lb_a = backend.elbv2.create_load_balancer("mh-a", region="us-east-1")
    lb_b = backend.elbv2.create_load_balancer("mh-b", region="us-east-1")
    svc = corev1.Service(
        metadata=ObjectMeta(name="multi", namespace="default"),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=80, protocol="TCP")],
        ),
    )
    for _ in range(3):  # stable across reconciles, not oscillating
        for lb, name in ((lb_a, "mh-a"), (lb_b, "mh-b")):
            cloud.ensure_global_accelerator_for_service(
                svc, corev1.LoadBalancerIngress(hostname=lb.dns_name),
                "c", name, "us-east-1",
            )
        accs, _ = backend.ga.list_accelerators()
        assert len(accs) == 1
        tags = {t.key: t.value for t in
                backend.ga.list_tags_for_resource(accs[0].accelerator_arn)}
        assert tags["aws-global-accelerator-target-hostname"] == lb_b.dns_name
