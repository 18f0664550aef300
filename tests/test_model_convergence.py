"""Model-based convergence: a seeded random walk of user operations
(create / annotate / de-annotate / change ports / change hostname /
delete) against the full three-controller stack, then a full settle, then
an exact audit of the cloud against the desired-state model:

- every live managed Service with an active LB owns EXACTLY one
  accelerator, tagged with its hostname, with one listener carrying the
  service's ports and one endpoint group containing the LB;
- no accelerator exists for deleted or unmanaged services;
- every live route53-hostname annotation owns exactly its TXT + A pair,
  and no owned records exist for dropped hostnames.

This is the controller-tier analogue of a numerics check: instead of
comparing kernels against an fp32 reference, the converged cloud is
compared against an independent model of what the annotations demand.
"""

import random
import threading
import time

import pytest

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws.client import FakeCloudFactory
from agac.cloudprovider.fake import FakeAWSBackend
from agac.kube.client import InMemoryKubeClient
from agac.kube.store import NotFoundError
from agac.manager import ControllerConfig, Manager

REGION = "us-east-1"
MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"
OWNER_TAG = "aws-global-accelerator-owner"
HOSTNAME_TAG = "aws-global-accelerator-target-hostname"


class Model:
    """The desired state implied by the current k8s objects, plus the
    path-dependent route53 bounds: changing a hostname annotation leaves
    the old records behind (reference ensure only processes CURRENT
    hostnames), and records created while managed survive un-managing —
    only hostname-annotation removal or object deletion cleans up
    (reference r53/service.go:29-111).  So the audit checks
    must ⊆ live-A-records ⊆ maybe instead of exact equality."""

    def __init__(self):
        self.services = {}  # name -> {"managed": bool, "ports": [..], "hostname": str|None}
        self.maybe_hostnames = {}  # name -> set of hostnames possibly recorded

    def managed(self):
        return {n: s for n, s in self.services.items() if s["managed"]}


def settle(predicate, timeout=45.0, what="convergence"):
    deadline = time.monotonic() + timeout
    last_err = None
    while time.monotonic() < deadline:
        try:
            if predicate():
                return
        except Exception as e:  # audit raced a mutation; retry
            last_err = e
        time.sleep(0.05)
    raise TimeoutError(f"{what} not reached: {last_err}")


def run_walk(seed: int, n_services: int = 6, n_ops: int = 60, api: str = "memory"):
    rng = random.Random(seed)
    backend = FakeAWSBackend()
    server = None
    if api == "http":
        from agac.kube.httpapi import APIServer
        from agac.kube.k8s import K8sKubeClient
        from agac.kube.kubeconfig import RestConfig
        from agac.kube.store import APIStore

        server = APIServer(APIStore(), watch_idle_seconds=0.1)
        server.start()
        client = K8sKubeClient(RestConfig(host=server.url), page_size=3)
    else:
        client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    # fast GA-missing retry so route53 reconciles that race the GA
    # creation converge within the settle budget (the reference's 60s
    # default is injectable)
    factory = FakeCloudFactory(backend, ga_missing_retry=0.1)
    manager.run(client, ControllerConfig(), factory, stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    model = Model()
    backend.route53.create_hosted_zone("walk.example.com")
    lbs = {}

    def k8s_service(name):
        m = model.services[name]
        annotations = {LB_TYPE: "nlb"}
        if m["managed"]:
            annotations[MANAGED] = "true"
        if m["hostname"]:
            annotations[HOSTNAME] = m["hostname"]
        return corev1.Service(
            metadata=ObjectMeta(name=name, namespace="default",
                                annotations=annotations),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=p, protocol="TCP")
                       for p in m["ports"]],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(
                        hostname=lbs[name].dns_name)]
                )
            ),
        )

    def push(name):
        """Write the model's desired object to the API (create or update)."""
        desired = k8s_service(name)
        for attempt in range(20):
            try:
                live = client.get("Service", "default", name)
            except NotFoundError:
                client.create(desired)
                return
            live.metadata.annotations = desired.metadata.annotations
            live.spec.ports = desired.spec.ports
            try:
                client.update(live)
                return
            except Exception:
                continue
        raise RuntimeError(f"could not push {name}")

    try:
        for op_i in range(n_ops):
            name = f"walk-{rng.randrange(n_services)}"
            op = rng.choice(
                ["create", "manage", "unmanage", "ports", "hostname",
                 "drop_hostname", "delete"]
            )
            if name not in model.services:
                if op == "delete":
                    continue
                if name not in lbs:
                    lbs[name] = backend.elbv2.create_load_balancer(
                        name, region=REGION
                    )
                model.services[name] = {
                    "managed": False, "ports": [80], "hostname": None,
                }
                push(name)
                continue
            m = model.services[name]
            if op == "create":
                continue
            if op == "manage":
                m["managed"] = True
            elif op == "unmanage":
                m["managed"] = False
            elif op == "ports":
                m["ports"] = sorted(rng.sample(range(1000, 1010), rng.randint(1, 3)))
            elif op == "hostname":
                m["hostname"] = f"{name}-{rng.randrange(3)}.walk.example.com"
                model.maybe_hostnames.setdefault(name, set()).add(m["hostname"])
            elif op == "drop_hostname":
                # annotation removal cleans up every owned record
                m["hostname"] = None
                model.maybe_hostnames.pop(name, None)
            elif op == "delete":
                client.delete("Service", "default", name)
                del model.services[name]
                model.maybe_hostnames.pop(name, None)
                continue
            push(name)

        # ---- audit ------------------------------------------------------
        def audit():
            managed = model.managed()
            accs, _ = backend.ga.list_accelerators()
            owned = {}
            for a in accs:
                tags = {t.key: t.value for t in
                        backend.ga.list_tags_for_resource(a.accelerator_arn)}
                owner = tags.get(OWNER_TAG)
                if owner is None:
                    return False  # mid-create
                owned.setdefault(owner, []).append((a, tags))
            # exactly the managed set, one accelerator each
            want_owners = {f"service/default/{n}" for n in managed}
            if set(owned) != want_owners:
                return False
            for name, m in managed.items():
                entries = owned[f"service/default/{name}"]
                if len(entries) != 1:
                    return False
                acc, tags = entries[0]
                if tags.get(HOSTNAME_TAG) != lbs[name].dns_name:
                    return False
                listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
                if len(listeners) != 1:
                    return False
                got_ports = sorted(p.from_port for p in listeners[0].port_ranges)
                if got_ports != sorted(m["ports"]):
                    return False
                groups, _ = backend.ga.list_endpoint_groups(
                    listeners[0].listener_arn)
                if len(groups) != 1:
                    return False
                if not any(d.endpoint_id == lbs[name].load_balancer_arn
                           for d in groups[0].endpoint_descriptions):
                    return False
            # route53 bounds (see Model docstring): currently managed
            # services' hostnames MUST exist; nothing outside the
            # could-have-been-created set MAY exist
            zones = backend.route53._zones
            zone_id = next(iter(zones))
            recs, _ = backend.route53.list_resource_record_sets(zone_id)
            a_names = {r.name for r in recs if r.type == "A"}
            must = {
                m["hostname"] + "."
                for m in managed.values()
                if m["hostname"]
            }
            maybe = {
                h + "."
                for hs in model.maybe_hostnames.values()
                for h in hs
            }
            if not (must <= a_names <= maybe | must):
                return False
            return True

        settle(audit, what=f"seed {seed} walk audit")

        # mutation check: the audit must actually discriminate — corrupt
        # one converged listener and confirm the audit now fails (guards
        # against a vacuously-true audit)
        managed_now = model.managed()
        if managed_now:
            accs, _ = backend.ga.list_accelerators()
            if accs:
                listeners, _ = backend.ga.list_listeners(accs[0].accelerator_arn)
                if listeners:
                    from agac.cloudprovider.aws import types as awstypes

                    backend.ga.update_listener(
                        listeners[0].listener_arn,
                        port_ranges=[awstypes.PortRange(65000, 65000)],
                    )
                    assert not audit(), "audit blind to a corrupted listener"
    finally:
        stop.set()
        if server is not None:
            server.shutdown()


@pytest.mark.parametrize("seed", [7, 23, 1009, 31337, 77, 5150])
def test_random_walk_converges_to_model(seed):
    run_walk(seed)


def test_long_walk_converges_to_model():
    """A deeper walk: more services, more ops, more interleavings."""
    run_walk(999331, n_services=12, n_ops=200)


def test_random_walk_over_production_wire_client():
    """The same walk through K8sKubeClient -> HTTP (page_size=3 keeps the
    informer syncs multi-page)."""
    run_walk(4242, api="http")


# ---------------------------------------------------------------------------
# Ingress walk: the ALB-ingress paths through the same model audit
# ---------------------------------------------------------------------------
LISTEN_PORTS = "alb.ingress.kubernetes.io/listen-ports"


def run_ingress_walk(seed: int, n_ingresses: int = 5, n_ops: int = 50):
    rng = random.Random(seed)
    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    factory = FakeCloudFactory(backend, ga_missing_retry=0.1)
    manager.run(client, ControllerConfig(), factory, stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    model = {}  # name -> {"managed": bool, "ports": [..]}
    lbs = {}

    def k8s_ingress(name):
        m = model[name]
        annotations = {LISTEN_PORTS: str(
            [{"HTTP": p} for p in m["ports"]]).replace("'", '"')}
        if m["managed"]:
            annotations[MANAGED] = "true"
        return corev1.Ingress(
            metadata=ObjectMeta(name=name, namespace="default",
                                annotations=annotations),
            spec=corev1.IngressSpec(ingress_class_name="alb"),
            status=corev1.IngressStatus(
                load_balancer=corev1.IngressLoadBalancerStatus(
                    ingress=[corev1.IngressLoadBalancerIngress(
                        hostname=lbs[name].dns_name)]
                )
            ),
        )

    def push(name):
        desired = k8s_ingress(name)
        for _ in range(20):
            try:
                live = client.get("Ingress", "default", name)
            except NotFoundError:
                client.create(desired)
                return
            live.metadata.annotations = desired.metadata.annotations
            try:
                client.update(live)
                return
            except Exception:
                continue

    try:
        for _ in range(n_ops):
            name = f"ing-{rng.randrange(n_ingresses)}"
            op = rng.choice(["create", "manage", "unmanage", "ports", "delete"])
            if name not in model:
                if op == "delete":
                    continue
                if name not in lbs:
                    # ALB hostname shape so the parser takes the ALB branch
                    lbs[name] = backend.elbv2.create_load_balancer(
                        name, region=REGION, lb_type="application",
                    )
                model[name] = {"managed": False, "ports": [80]}
                push(name)
                continue
            if op == "create":
                continue
            if op == "manage":
                model[name]["managed"] = True
            elif op == "unmanage":
                model[name]["managed"] = False
            elif op == "ports":
                model[name]["ports"] = sorted(
                    rng.sample([80, 443, 8080, 8443, 9090], rng.randint(1, 3))
                )
            elif op == "delete":
                client.delete("Ingress", "default", name)
                del model[name]
                continue
            push(name)

        def audit():
            managed = {n: m for n, m in model.items() if m["managed"]}
            accs, _ = backend.ga.list_accelerators()
            owned = {}
            for a in accs:
                tags = {t.key: t.value for t in
                        backend.ga.list_tags_for_resource(a.accelerator_arn)}
                owner = tags.get(OWNER_TAG)
                if owner is None:
                    return False
                owned.setdefault(owner, []).append(a)
            if set(owned) != {f"ingress/default/{n}" for n in managed}:
                return False
            for name, m in managed.items():
                entries = owned[f"ingress/default/{name}"]
                if len(entries) != 1:
                    return False
                listeners, _ = backend.ga.list_listeners(entries[0].accelerator_arn)
                if len(listeners) != 1:
                    return False
                if listeners[0].protocol != "TCP":  # ALB listeners are TCP
                    return False
                if sorted(p.from_port for p in listeners[0].port_ranges) != m["ports"]:
                    return False
            return True

        settle(audit, what=f"ingress walk seed {seed}")
    finally:
        stop.set()


@pytest.mark.parametrize("seed", [11, 4242, 90210])
def test_ingress_random_walk_converges(seed):
    run_ingress_walk(seed)


# ---------------------------------------------------------------------------
# Walks under fault injection: throttles during churn, exact audit after
# ---------------------------------------------------------------------------
def run_faulty_walk(seed: int, n_services: int = 6, n_ops: int = 50,
                    fault_rate: float = 0.25, target_ops=None):
    """Same walk, but while ops are flowing every AWS call has a
    ``fault_rate`` chance of throwing ThrottlingException.  Faults stop
    before settle; the audit must still converge exactly — proving the
    rate-limited retry paths lose nothing."""
    from agac.cloudprovider.aws import errors as awserr

    rng = random.Random(seed)
    fault_rng = random.Random(seed ^ 0xFA17)
    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    factory = FakeCloudFactory(backend, ga_missing_retry=0.1)
    # bound the per-item failure backoff (client-go knob, configurable
    # since r2): under a 25% per-call fault rate a reconcile fails ~80% of
    # the time, so the default 1000s cap would legitimately delay a single
    # item's recovery by minutes after the storm — irrelevant to what this
    # test verifies (no LOST updates)
    from agac.controller.endpointgroupbinding import EndpointGroupBindingConfig
    from agac.controller.globalaccelerator import GlobalAcceleratorConfig
    from agac.controller.route53 import Route53Config

    config = ControllerConfig(
        global_accelerator=GlobalAcceleratorConfig(queue_item_max_delay=0.5),
        route53=Route53Config(queue_item_max_delay=0.5),
        endpoint_group_binding=EndpointGroupBindingConfig(queue_item_max_delay=0.5),
    )
    manager.run(client, config, factory, stop,
                resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    model = Model()
    backend.route53.create_hosted_zone("walk.example.com")
    lbs = {}
    faults = {"n": 0}

    def hook(service, op):
        if target_ops is not None and op not in target_ops:
            return
        if fault_rng.random() < fault_rate:
            faults["n"] += 1
            raise awserr.AWSAPIError("injected", "ThrottlingException")

    backend.set_fault_hook(hook)

    def k8s_service(name):
        m = model.services[name]
        annotations = {LB_TYPE: "nlb"}
        if m["managed"]:
            annotations[MANAGED] = "true"
        if m["hostname"]:
            annotations[HOSTNAME] = m["hostname"]
        return corev1.Service(
            metadata=ObjectMeta(name=name, namespace="default",
                                annotations=annotations),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=p, protocol="TCP")
                       for p in m["ports"]],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(
                        hostname=lbs[name].dns_name)]
                )
            ),
        )

    def push(name):
        desired = k8s_service(name)
        for _ in range(20):
            try:
                live = client.get("Service", "default", name)
            except NotFoundError:
                client.create(desired)
                return
            live.metadata.annotations = desired.metadata.annotations
            live.spec.ports = desired.spec.ports
            try:
                client.update(live)
                return
            except Exception:
                continue

    try:
        for _ in range(n_ops):
            name = f"walk-{rng.randrange(n_services)}"
            op = rng.choice(["create", "manage", "unmanage", "ports",
                             "hostname", "drop_hostname", "delete"])
            if name not in model.services:
                if op == "delete":
                    continue
                if name not in lbs:
                    # LB creation is test setup, not a controller call —
                    # bypass the hook for it
                    backend.elbv2.fault_hook = None
                    lbs[name] = backend.elbv2.create_load_balancer(
                        name, region=REGION)
                    backend.elbv2.fault_hook = hook
                model.services[name] = {
                    "managed": False, "ports": [80], "hostname": None}
                push(name)
                continue
            m = model.services[name]
            if op == "create":
                continue
            if op == "manage":
                m["managed"] = True
            elif op == "unmanage":
                m["managed"] = False
            elif op == "ports":
                m["ports"] = sorted(rng.sample(range(1000, 1010),
                                               rng.randint(1, 3)))
            elif op == "hostname":
                m["hostname"] = f"{name}-{rng.randrange(3)}.walk.example.com"
                model.maybe_hostnames.setdefault(name, set()).add(m["hostname"])
            elif op == "drop_hostname":
                m["hostname"] = None
                model.maybe_hostnames.pop(name, None)
            elif op == "delete":
                client.delete("Service", "default", name)
                del model.services[name]
                model.maybe_hostnames.pop(name, None)
                continue
            push(name)
            time.sleep(0.005)  # let reconciles interleave with faults

        backend.set_fault_hook(None)  # faults clear; now converge exactly
        if target_ops is None:
            assert faults["n"] > 0, "fault injection never fired"

        def audit():
            managed = model.managed()
            accs, _ = backend.ga.list_accelerators()
            owned = {}
            for a in accs:
                tags = {t.key: t.value for t in
                        backend.ga.list_tags_for_resource(a.accelerator_arn)}
                owner = tags.get(OWNER_TAG)
                if owner is None:
                    return False
                owned.setdefault(owner, []).append((a, tags))
            if set(owned) != {f"service/default/{n}" for n in managed}:
                return False
            for name, m in managed.items():
                entries = owned[f"service/default/{name}"]
                if len(entries) != 1:
                    return False
                acc, tags = entries[0]
                if tags.get(HOSTNAME_TAG) != lbs[name].dns_name:
                    return False
                listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
                if len(listeners) != 1:
                    return False
                if sorted(p.from_port for p in listeners[0].port_ranges) != \
                        sorted(m["ports"]):
                    return False
                groups, _ = backend.ga.list_endpoint_groups(
                    listeners[0].listener_arn)
                if len(groups) != 1:
                    return False
            return True

        settle(audit, timeout=60.0, what=f"faulty walk seed {seed}")
    finally:
        backend.set_fault_hook(None)
        stop.set()


@pytest.mark.parametrize("seed", [3, 666, 80486])
def test_faulty_walk_converges_after_faults_clear(seed):
    run_faulty_walk(seed)


OP_FAMILIES = {
    "deletes": {"delete_accelerator", "delete_listener", "delete_endpoint_group"},
    "creates": {"create_accelerator", "create_listener", "create_endpoint_group"},
    "describes": {"describe_accelerator", "describe_load_balancers",
                  "describe_endpoint_group", "list_listeners",
                  "list_endpoint_groups", "list_accelerators"},
    "route53": {"change_resource_record_sets", "list_resource_record_sets",
                "list_hosted_zones_by_name", "list_hosted_zones"},
}


@pytest.mark.parametrize("family", sorted(OP_FAMILIES))
def test_concentrated_storm_on_one_op_family_converges(family):
    """70% fault rate concentrated on a single op family (the shape that
    exposed the PARITY §5c cleanup leak on describes): every family must
    converge exactly once the storm clears.  Swept 60 seeds per family
    offline; one representative seed pinned here."""
    run_faulty_walk(400007, n_services=6, n_ops=50, fault_rate=0.7,
                    target_ops=OP_FAMILIES[family])


def run_drift_walk(seed: int, n_services: int = 6, n_ops: int = 60):
    """Walks with an extra adversary: ops that corrupt the CLOUD directly
    (listener ports, tags) behind the controllers' backs.  With
    cloud-resync enabled, the audit must converge to the model anyway —
    the sustained version of tests/test_cloud_resync.py."""
    rng = random.Random(seed)
    backend = FakeAWSBackend()
    client = InMemoryKubeClient()
    stop = threading.Event()
    manager = Manager()
    from agac.controller.endpointgroupbinding import EndpointGroupBindingConfig
    from agac.controller.globalaccelerator import GlobalAcceleratorConfig
    from agac.controller.route53 import Route53Config

    config = ControllerConfig(
        global_accelerator=GlobalAcceleratorConfig(cloud_resync_period=0.2),
        route53=Route53Config(cloud_resync_period=0.2),
        endpoint_group_binding=EndpointGroupBindingConfig(cloud_resync_period=0.2),
    )
    factory = FakeCloudFactory(backend, ga_missing_retry=0.1)
    manager.run(client, config, factory, stop, resync_period=300.0, block=False)
    assert manager.wait_until_ready()
    model = Model()
    backend.route53.create_hosted_zone("walk.example.com")
    lbs = {}

    def k8s_service(name):
        m = model.services[name]
        annotations = {LB_TYPE: "nlb"}
        if m["managed"]:
            annotations[MANAGED] = "true"
        return corev1.Service(
            metadata=ObjectMeta(name=name, namespace="default",
                                annotations=annotations),
            spec=corev1.ServiceSpec(
                type="LoadBalancer",
                ports=[corev1.ServicePort(port=p, protocol="TCP")
                       for p in m["ports"]],
            ),
            status=corev1.ServiceStatus(
                load_balancer=corev1.LoadBalancerStatus(
                    ingress=[corev1.LoadBalancerIngress(
                        hostname=lbs[name].dns_name)]
                )
            ),
        )

    def push(name):
        desired = k8s_service(name)
        for _ in range(20):
            try:
                live = client.get("Service", "default", name)
            except NotFoundError:
                client.create(desired)
                return
            live.metadata.annotations = desired.metadata.annotations
            live.spec.ports = desired.spec.ports
            try:
                client.update(live)
                return
            except Exception:
                continue

    def corrupt_cloud():
        """Mutate a random accelerator's listener or tags directly."""
        accs, _ = backend.ga.list_accelerators()
        if not accs:
            return
        acc = rng.choice(accs)
        listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
        if listeners and rng.random() < 0.7:
            from agac.cloudprovider.aws import types as t

            backend.ga.update_listener(
                listeners[0].listener_arn,
                port_ranges=[t.PortRange(65000 + rng.randrange(100), 65000)],
            )
        else:
            from agac.cloudprovider.aws import types as t

            backend.ga.tag_resource(acc.accelerator_arn, [
                t.Tag("aws-global-accelerator-target-hostname", "corrupted")
            ])

    try:
        for _ in range(n_ops):
            op = rng.choice(["create", "manage", "unmanage", "ports",
                             "delete", "corrupt", "corrupt"])
            if op == "corrupt":
                corrupt_cloud()
                continue
            name = f"walk-{rng.randrange(n_services)}"
            if name not in model.services:
                if op == "delete":
                    continue
                if name not in lbs:
                    lbs[name] = backend.elbv2.create_load_balancer(
                        name, region=REGION)
                model.services[name] = {
                    "managed": False, "ports": [80], "hostname": None}
                push(name)
                continue
            m = model.services[name]
            if op == "create":
                continue
            if op == "manage":
                m["managed"] = True
            elif op == "unmanage":
                m["managed"] = False
            elif op == "ports":
                m["ports"] = sorted(rng.sample(range(1000, 1010),
                                               rng.randint(1, 3)))
            elif op == "delete":
                client.delete("Service", "default", name)
                del model.services[name]
                continue
            push(name)

        def audit():
            managed = model.managed()
            accs, _ = backend.ga.list_accelerators()
            owned = {}
            for a in accs:
                tags = {t.key: t.value for t in
                        backend.ga.list_tags_for_resource(a.accelerator_arn)}
                owner = tags.get(OWNER_TAG)
                if owner is None:
                    return False
                owned.setdefault(owner, []).append((a, tags))
            if set(owned) != {f"service/default/{n}" for n in managed}:
                return False
            for name, m in managed.items():
                entries = owned[f"service/default/{name}"]
                if len(entries) != 1:
                    return False
                acc, tags = entries[0]
                if tags.get(HOSTNAME_TAG) != lbs[name].dns_name:
                    return False  # corrupted tag must be repaired
                listeners, _ = backend.ga.list_listeners(acc.accelerator_arn)
                if len(listeners) != 1:
                    return False
                if sorted(p.from_port for p in listeners[0].port_ranges) != \
                        sorted(m["ports"]):
                    return False  # corrupted ports must be repaired
            return True

        settle(audit, timeout=30.0, what=f"drift walk seed {seed}")
    finally:
        stop.set()


@pytest.mark.parametrize("seed", [55, 808, 31415])
def test_drift_walk_self_heals_with_cloud_resync(seed):
    run_drift_walk(seed)
