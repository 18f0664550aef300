"""AWS-side fault injection: transient 5xx and
sustained throttling injected into the fake AWS via
``FakeAWSBackend.set_fault_hook`` prove that

- the partial-create rollback (reference global_accelerator.go:142-147)
  leaves no orphaned accelerator/listener after a mid-triple failure,
- the rate-limited requeue retries to convergence under sustained
  throttling (reconcile.go:70-76 semantics),
- the Route53 TXT+A pair-create converges after a mid-pair failure
  (deliberate fix, docs/PARITY.md §5b — the reference would loop on
  InvalidChangeBatch forever).
"""

import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import errors as awserr
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.client import InMemoryKubeClient
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"


class Throttle(awserr.AWSAPIError):
    code = "ThrottlingException"


class InternalError(awserr.AWSAPIError):
    code = "InternalServiceErrorException"


def managed_service(name, lb, hostname_annotation=None):
    annotations = {
        "service.beta.kubernetes.io/aws-load-balancer-type": "nlb",
        "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
    }
    if hostname_annotation:
        annotations[
            "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
        ] = hostname_annotation
    return corev1.Service(
        metadata=ObjectMeta(name=name, namespace="default", annotations=annotations),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=80, protocol="TCP")],
        ),
        status=corev1.ServiceStatus(
            load_balancer=corev1.LoadBalancerStatus(
                ingress=[corev1.LoadBalancerIngress(hostname=lb.dns_name)]
            )
        ),
    )


def start_stack(backend):
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    manager.run(client, ControllerConfig(), FakeCloudFactory(backend), stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    return client, stop


def wait_for(predicate, what, timeout=30.0):
    deadline = time.monotonic() + timeout
    while not predicate():
        if time.monotonic() > deadline:
            raise TimeoutError(f"{what} within {timeout}s")
        time.sleep(0.02)


def triple_counts(backend):
    accs, _ = backend.ga.list_accelerators()
    n_listeners = sum(
        len(backend.ga.list_listeners(a.accelerator_arn)[0]) for a in accs
    )
    return len(accs), n_listeners


class TestPartialCreateRollback:
    def test_endpoint_group_5xx_rolls_back_then_converges(self):
        """create_endpoint_group fails twice with a 5xx: each failed
        attempt must roll the accelerator+listener back (no orphans), and
        the rate-limited retry must converge once the fault clears."""
        backend = FakeAWSBackend()
        fails = {"n": 0}
        orphan_snapshots = []

        def hook(service, op):
            if service == "globalaccelerator" and op == "create_endpoint_group":
                if fails["n"] < 2:
                    fails["n"] += 1
                    raise InternalError("injected 5xx")

        backend.set_fault_hook(hook)
        client, stop = start_stack(backend)
        try:
            lb = backend.elbv2.create_load_balancer("rollback", region=REGION)
            client.create(managed_service("rollback", lb))

            # wait until both injected failures have fired
            wait_for(lambda: fails["n"] == 2, "two injected failures")
            # any state observed between attempts must show no orphaned
            # accelerator-without-endpoint-group surviving: poll a few times
            for _ in range(10):
                accs, _ = backend.ga.list_accelerators()
                for a in accs:
                    listeners, _ = backend.ga.list_listeners(a.accelerator_arn)
                    for l in listeners:
                        groups, _ = backend.ga.list_endpoint_groups(l.listener_arn)
                        # a listener may transiently exist mid-create; an
                        # accelerator that SURVIVES with no endpoint group
                        # after rollback would stay orphaned forever, which
                        # the converged assertion below would catch
                orphan_snapshots.append(triple_counts(backend))
                time.sleep(0.02)

            # convergence: exactly one full triple, nothing orphaned
            def converged():
                accs, _ = backend.ga.list_accelerators()
                if len(accs) != 1:
                    return False
                listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
                if len(listeners) != 1:
                    return False
                groups, _ = backend.ga.list_endpoint_groups(listeners[0].listener_arn)
                return len(groups) == 1

            wait_for(converged, "triple after fault clears")
        finally:
            stop.set()

    def test_listener_5xx_never_leaks_accelerator(self):
        """A failure at create_listener must leave ZERO accelerators after
        rollback (the cleanup path), not a headless accelerator."""
        backend = FakeAWSBackend()
        state = {"fail": True, "saw_zero_after_fail": False}

        def hook(service, op):
            if (
                state["fail"]
                and service == "globalaccelerator"
                and op == "create_listener"
            ):
                raise InternalError("injected 5xx")

        backend.set_fault_hook(hook)
        client, stop = start_stack(backend)
        try:
            lb = backend.elbv2.create_load_balancer("leak", region=REGION)
            client.create(managed_service("leak", lb))
            # give the controller a few failed attempts
            wait_for(
                lambda: backend.ga.call_counts.get("create_listener", 0) >= 2,
                "repeated create attempts",
            )
            # rollback happened every time: no accelerator persists while
            # the fault is active (poll to dodge the in-flight window)
            def no_survivor():
                accs, _ = backend.ga.list_accelerators()
                return len(accs) == 0

            for _ in range(20):
                if no_survivor():
                    state["saw_zero_after_fail"] = True
                time.sleep(0.01)
            assert state["saw_zero_after_fail"], "rolled-back state never observed"
            state["fail"] = False
            wait_for(
                lambda: len(backend.ga.list_accelerators()[0]) == 1,
                "create after fault clears",
            )
        finally:
            stop.set()


class TestSustainedThrottling:
    def test_converges_through_throttling(self):
        """Every 3rd AWS call (any service) throttles for the first few
        seconds; all services must still converge, exactly once each."""
        backend = FakeAWSBackend()
        counter = {"n": 0, "throttling": True, "injected": 0}
        lock = threading.Lock()

        def hook(service, op):
            with lock:
                if not counter["throttling"]:
                    return
                counter["n"] += 1
                if counter["n"] % 3 == 0:
                    counter["injected"] += 1
                    raise Throttle(f"injected throttle for {service}.{op}")

        backend.set_fault_hook(hook)
        client, stop = start_stack(backend)
        try:
            lbs = [
                backend.elbv2.create_load_balancer(f"thr-{i}", region=REGION)
                for i in range(4)
            ]
            for i, lb in enumerate(lbs):
                client.create(managed_service(f"thr-{i}", lb))
            time.sleep(1.0)  # let retries grind against the throttle
            with lock:
                counter["throttling"] = False
            assert counter["injected"] > 0, "no faults were actually injected"

            def converged():
                accs, _ = backend.ga.list_accelerators()
                owners = set()
                for a in accs:
                    tags = backend.ga.list_tags_for_resource(a.accelerator_arn)
                    owners.add(
                        {t.key: t.value for t in tags}.get("aws-global-accelerator-owner")
                    )
                return len(accs) == 4 and len(owners) == 4

            wait_for(converged, "all services after throttling clears")
        finally:
            stop.set()


class TestRoute53PartialPair:
    def test_mid_pair_failure_converges(self):
        """TXT commits, A-record change fails transiently: the retry must
        converge the pair (PARITY §5b fix; the reference loops forever on
        InvalidChangeBatch here).  Driven directly through the resource
        manager so the failure lands deterministically on the second
        change batch of the ensure (the A record)."""
        backend = FakeAWSBackend()
        cloud = FakeCloudFactory(backend)(REGION)
        zone = backend.route53.create_hosted_zone("pair.example.com")
        lb = backend.elbv2.create_load_balancer("pair", region=REGION)
        svc = managed_service("pair", lb)
        arn, created, _ = cloud.ensure_global_accelerator_for_service(
            svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "default",
            "pair", REGION,
        )
        assert created

        state = {"changes": 0}

        def hook(service, op):
            if service == "route53" and op == "change_resource_record_sets":
                state["changes"] += 1
                if state["changes"] == 2:  # TXT committed; fail the A create
                    raise InternalError("injected 5xx mid-pair")

        backend.set_fault_hook(hook)
        with pytest.raises(awserr.AWSAPIError):
            cloud.ensure_route53_for_service(
                svc, corev1.LoadBalancerIngress(hostname=lb.dns_name),
                ["app.pair.example.com"], "default",
            )
        recs, _ = backend.route53.list_resource_record_sets(zone.id)
        kinds = {(r.name, r.type) for r in recs}
        assert ("app.pair.example.com.", "TXT") in kinds  # partial state
        assert ("app.pair.example.com.", "A") not in kinds

        backend.set_fault_hook(None)  # fault clears; the retry must converge
        cloud.ensure_route53_for_service(
            svc, corev1.LoadBalancerIngress(hostname=lb.dns_name),
            ["app.pair.example.com"], "default",
        )
        recs, _ = backend.route53.list_resource_record_sets(zone.id)
        kinds = {(r.name, r.type) for r in recs}
        assert {("app.pair.example.com.", "A"),
                ("app.pair.example.com.", "TXT")} <= kinds

    def test_txt_already_owned_is_idempotent_but_foreign_is_not(self):
        """Direct unit check of the recovery rule: our TXT ⇒ idempotent
        success; foreign TXT ⇒ InvalidChangeBatch still raised."""
        from agac.cloudprovider.aws.route53 import route53_owner_value

        backend = FakeAWSBackend()
        cloud = FakeCloudFactory(backend)(REGION)
        zone = backend.route53.create_hosted_zone("own.example.com")
        owner = route53_owner_value("c", "service", "default", "web")

        import agac.cloudprovider.aws.types as t

        # our own TXT pre-exists (simulated partial create)
        backend.route53.change_resource_record_sets(zone.id, [t.Change(
            action="CREATE",
            record_set=t.ResourceRecordSet(
                name="a.own.example.com.", type="TXT", ttl=300,
                resource_records=[t.ResourceRecord(value=owner)],
            ),
        )])
        stored_zone = t.HostedZone(id=zone.id, name=zone.name)
        cloud._create_metadata_record_set(stored_zone, "a.own.example.com", owner)  # no raise

        # a FOREIGN TXT must still raise
        backend.route53.change_resource_record_sets(zone.id, [t.Change(
            action="CREATE",
            record_set=t.ResourceRecordSet(
                name="b.own.example.com.", type="TXT", ttl=300,
                resource_records=[t.ResourceRecord(value="someone-else")],
            ),
        )])
        with pytest.raises(awserr.InvalidChangeBatch):
            cloud._create_metadata_record_set(stored_zone, "b.own.example.com", owner)


class TestCleanupLeakUnderThrottle:
    """PARITY §5c: one throttled DescribeAccelerator during teardown made
    cleanup report success and leak the accelerator forever (the reference
    shares this: listRelatedGlobalAccelerator treats ANY error as
    "already deleted").  Now transient errors propagate so the cleanup is
    retried to completion."""

    def test_throttled_discovery_does_not_fake_success(self):
        backend = FakeAWSBackend()
        cloud = FakeCloudFactory(backend)(REGION)
        lb = backend.elbv2.create_load_balancer("leak", region=REGION)
        svc = managed_service("leak", lb)
        arn, created, _ = cloud.ensure_global_accelerator_for_service(
            svc, corev1.LoadBalancerIngress(hostname=lb.dns_name), "c",
            "leak", REGION,
        )
        assert created

        def hook(service, op):
            if op == "describe_accelerator":
                raise Throttle("injected during teardown discovery")

        backend.set_fault_hook(hook)
        with pytest.raises(awserr.AWSAPIError):
            cloud.cleanup_global_accelerator(arn)  # must NOT fake success
        # accelerator still there (nothing silently forgotten)
        backend.set_fault_hook(None)
        assert len(backend.ga.list_accelerators()[0]) == 1
        # fault clears; the retried cleanup completes
        cloud.cleanup_global_accelerator(arn)
        assert backend.ga.list_accelerators()[0] == []

    def test_controller_retries_cleanup_to_completion(self):
        """End to end: unmanage during a throttle storm on describe; the
        rate-limited requeue must eventually delete the accelerator
        instead of leaking it."""
        backend = FakeAWSBackend()
        client, stop = start_stack(backend)
        try:
            lb = backend.elbv2.create_load_balancer("ctl", region=REGION)
            client.create(managed_service("ctl", lb))
            wait_for(lambda: len(backend.ga.list_accelerators()[0]) == 1, "create")

            state = {"fail": True}

            def hook(service, op):
                if state["fail"] and op == "describe_accelerator":
                    raise Throttle("storm")

            backend.set_fault_hook(hook)
            svc = client.get("Service", "default", "ctl")
            del svc.metadata.annotations[
                "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
            ]
            client.update(svc)
            time.sleep(0.5)  # cleanup attempts keep failing (and retrying)
            assert len(backend.ga.list_accelerators()[0]) == 1
            state["fail"] = False
            wait_for(lambda: len(backend.ga.list_accelerators()[0]) == 0,
                     "retried cleanup after storm clears")
        finally:
            stop.set()
