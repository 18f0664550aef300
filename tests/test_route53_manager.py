"""Route53 resource-manager flows against the fake: ensure (create/upsert/
skip), hosted-zone parent walk, TXT ownership, wildcard handling, cleanup."""

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.aws.route53 import route53_owner_value
from agac.cloudprovider.fake import FakeAWSBackend

REGION = "us-east-1"
CLUSTER = "c1"


@pytest.fixture
def backend():
    return FakeAWSBackend()


@pytest.fixture
def cloud(backend):
    return FakeCloudFactory(backend)(REGION)


def mk_service(name="web", ns="default"):
    return corev1.Service(metadata=ObjectMeta(name=name, namespace=ns))


def seed_ga_for_lb(backend, cloud, lb):
    """Create a managed accelerator triple owned by service/default/web."""
    svc = mk_service()
    arn, _, _ = cloud.ensure_global_accelerator_for_service(
        svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), CLUSTER, lb.load_balancer_name, REGION
    )
    return backend.ga.describe_accelerator(arn)


def records(backend, zone):
    recs, _ = backend.route53.list_resource_record_sets(zone.id)
    return {(r.name, r.type): r for r in recs}


class TestEnsureRoute53:
    def test_creates_txt_and_alias_pair(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        acc = seed_ga_for_lb(backend, cloud, lb)
        created, retry = cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        assert created and retry == 0
        recs = records(backend, zone)
        txt = recs[("www.example.com.", "TXT")]
        assert txt.ttl == 300
        assert txt.resource_records[0].value == route53_owner_value(
            CLUSTER, "service", "default", "web"
        )
        alias = recs[("www.example.com.", "A")]
        assert alias.alias_target.dns_name == acc.dns_name + "."
        assert alias.alias_target.hosted_zone_id == "Z2BJ6XQ5FK7U4H"
        assert alias.alias_target.evaluate_target_health is True

    def test_idempotent_skip_when_synced(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        args = (
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        created1, _ = cloud.ensure_route53_for_service(*args)
        created2, _ = cloud.ensure_route53_for_service(*args)
        assert created1 and not created2
        assert len(records(backend, zone)) == 2

    def test_upsert_on_ga_dns_drift(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        acc = seed_ga_for_lb(backend, cloud, lb)
        cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        # simulate stale alias pointing at an old accelerator DNS
        backend.route53.change_resource_record_sets(
            zone.id,
            [
                t.Change(
                    action=t.CHANGE_ACTION_UPSERT,
                    record_set=t.ResourceRecordSet(
                        name="www.example.com",
                        type=t.RR_TYPE_A,
                        alias_target=t.AliasTarget(
                            dns_name="old.awsglobalaccelerator.com.",
                            hosted_zone_id="Z2BJ6XQ5FK7U4H",
                        ),
                    ),
                )
            ],
        )
        created, _ = cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        assert not created
        alias = records(backend, zone)[("www.example.com.", "A")]
        assert alias.alias_target.dns_name == acc.dns_name + "."

    def test_no_ga_requeues_60s(self, backend, cloud):
        backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        created, retry = cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        assert not created and retry == 60.0

    def test_multiple_hostnames(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        created, _ = cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com", "api.example.com"],
            CLUSTER,
        )
        assert created
        recs = records(backend, zone)
        assert ("www.example.com.", "A") in recs
        assert ("api.example.com.", "A") in recs
        assert len(recs) == 4

    def test_wildcard_hostname_roundtrip(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        args = (
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["*.example.com"],
            CLUSTER,
        )
        created, _ = cloud.ensure_route53_for_service(*args)
        assert created
        # stored octal-escaped, and a second ensure matches it (no dup create)
        assert ("\\052.example.com.", "A") in records(backend, zone)
        created2, _ = cloud.ensure_route53_for_service(*args)
        assert not created2

    def test_no_hosted_zone_errors(self, backend, cloud):
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        with pytest.raises(ValueError, match="Could not find hosted zone"):
            cloud.ensure_route53_for_service(
                mk_service(),
                corev1.LoadBalancerIngress(hostname=lb.dns_name),
                ["www.nozone.net"],
                CLUSTER,
            )


class TestHostedZoneWalk:
    def test_walks_to_parent_zone(self, backend, cloud):
        backend.route53.create_hosted_zone("example.com")
        zone = cloud.get_hosted_zone("a.b.example.com")
        assert zone.name == "example.com."

    def test_prefers_most_specific(self, backend, cloud):
        backend.route53.create_hosted_zone("example.com")
        backend.route53.create_hosted_zone("sub.example.com")
        zone = cloud.get_hosted_zone("www.sub.example.com")
        assert zone.name == "sub.example.com."


class TestCleanup:
    def test_cleanup_removes_owned_records_only(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"],
            CLUSTER,
        )
        # foreign records must survive
        backend.route53.change_resource_record_sets(
            zone.id,
            [
                t.Change(
                    action=t.CHANGE_ACTION_CREATE,
                    record_set=t.ResourceRecordSet(
                        name="other.example.com",
                        type=t.RR_TYPE_A,
                        alias_target=t.AliasTarget(dns_name="x.", hosted_zone_id="Z"),
                    ),
                )
            ],
        )
        cloud.cleanup_record_set(CLUSTER, "service", "default", "web")
        recs = records(backend, zone)
        assert list(recs) == [("other.example.com.", "A")]

    def test_cleanup_is_noop_without_records(self, backend, cloud):
        backend.route53.create_hosted_zone("example.com")
        cloud.cleanup_record_set(CLUSTER, "service", "default", "ghost")

    def test_cleanup_scans_all_zones(self, backend, cloud):
        z1 = backend.route53.create_hosted_zone("one.com")
        z2 = backend.route53.create_hosted_zone("two.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        cloud.ensure_route53_for_service(
            mk_service(),
            corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["a.one.com", "b.two.com"],
            CLUSTER,
        )
        cloud.cleanup_record_set(CLUSTER, "service", "default", "web")
        assert records(backend, z1) == {}
        assert records(backend, z2) == {}


class TestOwnedRecordDiscoveryTyping:
    """_find_owned_a_record_at must only treat type=A alias
    records as the managed record (reference findARecord filters RRTypeA,
    route53.go:360-367) — a pre-existing AAAA alias at the same name must
    not be mistaken for it."""

    def test_aaaa_alias_at_name_is_not_the_managed_record(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        # first ensure creates TXT + A-alias
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER,
        )
        # someone adds an AAAA alias at the same name pointing elsewhere
        backend.route53.change_resource_record_sets(zone.id, [t.Change(
            action="CREATE",
            record_set=t.ResourceRecordSet(
                name="www.example.com.", type="AAAA",
                alias_target=t.AliasTarget(
                    hosted_zone_id="Z2BJ6XQ5FK7U4H",
                    dns_name="stale.awsglobalaccelerator.com",
                    evaluate_target_health=True,
                ),
            ),
        )])
        # second ensure: the managed A record is current → no upsert; the
        # AAAA record must be left alone (not taken as the A record and
        # "updated")
        created, retry = cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER,
        )
        assert not created and retry == 0
        recs = records(backend, zone)
        aaaa = recs[("www.example.com.", "AAAA")]
        assert aaaa.alias_target.dns_name.rstrip(".") == "stale.awsglobalaccelerator.com"
        assert ("www.example.com.", "A") in recs


class TestZoneHints:
    """Controller-held hostname→zone cache skips the parent-domain walk;
    stale hints self-heal via NoSuchHostedZone fallback and a TTL."""

    def _counting_hook(self, backend):
        counts = {}

        def hook(service, op):
            counts[(service, op)] = counts.get((service, op), 0) + 1

        backend.set_fault_hook(hook)
        return counts

    def test_hint_skips_zone_walk(self, backend, cloud):
        backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        hints = {}
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER, zone_hints=hints,
        )
        assert "www.example.com" in hints
        counts = self._counting_hook(backend)
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER, zone_hints=hints,
        )
        assert counts.get(("route53", "list_hosted_zones_by_name"), 0) == 0

    def test_stale_hint_falls_back_to_walk(self, backend, cloud):
        zone = backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        hints = {}
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER, zone_hints=hints,
        )
        # zone deleted and recreated under a new id
        del backend.route53._zones[zone.id]
        del backend.route53._records[zone.id]
        new_zone = backend.route53.create_hosted_zone("example.com")
        created, retry = cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.example.com"], CLUSTER, zone_hints=hints,
        )
        assert created and retry == 0
        recs, _ = backend.route53.list_resource_record_sets(new_zone.id)
        assert {("www.example.com.", "A"), ("www.example.com.", "TXT")} <= {
            (r.name, r.type) for r in recs
        }
        assert hints["www.example.com"][0].id == new_zone.id

    def test_hint_ttl_reruns_walk(self, backend, cloud, monkeypatch):
        from agac.cloudprovider.aws import route53 as r53mod

        backend.route53.create_hosted_zone("example.com")
        lb = backend.elbv2.create_load_balancer("mylb", region=REGION)
        seed_ga_for_lb(backend, cloud, lb)
        hints = {}
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.sub.example.com"], CLUSTER, zone_hints=hints,
        )
        # a more specific zone appears; the reference's per-reconcile walk
        # would pick it — the hint must honor it within one TTL
        specific = backend.route53.create_hosted_zone("sub.example.com")
        monkeypatch.setattr(r53mod, "ZONE_HINT_TTL", 0.0)
        cloud.ensure_route53_for_service(
            mk_service(), corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["www.sub.example.com"], CLUSTER, zone_hints=hints,
        )
        assert hints["www.sub.example.com"][0].id == specific.id
        recs, _ = backend.route53.list_resource_record_sets(specific.id)
        assert ("www.sub.example.com.", "A") in {(r.name, r.type) for r in recs}
