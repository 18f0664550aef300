"""Stateful in-memory AWS: ELBv2 + Global Accelerator + Route53.

The hermetic backend demanded by BASELINE.json ("mocked pkg/cloudprovider").
The reference has no AWS fake at all (its unit tests only cover pure
functions) — this fake models every AWS behavior the
controllers' retry/requeue logic depends on:

- typed error codes (LoadBalancerNotFound, AcceleratorNotFoundException,
  EndpointGroupNotFoundException, AcceleratorNotDisabledException, ...);
- pagination on every List* operation;
- load balancer ``provisioning → active`` state (drives the 30s requeue in
  the ensure paths);
- accelerator deploy lifecycle: every mutation sets IN_PROGRESS and the
  status flips to DEPLOYED after ``deploy_after_describes`` Describe calls
  (drives the disable→poll→delete loop);
- DeleteAccelerator requires the accelerator to be disabled and DEPLOYED;
- Route53 name normalization: record names are stored dot-terminated with
  ``*`` octal-escaped to ``\\052`` exactly as the real API returns them.

All three services share one lock, so cross-service invariants hold under
concurrent reconcile workers.
"""

from __future__ import annotations

import threading
import uuid
from typing import Dict, List, Optional, Tuple

from ..aws import errors as awserr
from ..aws import types as t

_ACCOUNT = "123456789012"


def _short() -> str:
    return uuid.uuid4().hex[:8]


def _paginate(items: list, max_results: Optional[int], token: Optional[str]):
    start = int(token) if token else 0
    if max_results is None or max_results <= 0:
        max_results = len(items) - start or 1
    page = items[start : start + max_results]
    next_token = str(start + max_results) if start + max_results < len(items) else None
    return page, next_token


class FakeELBv2:
    def __init__(self, lock: threading.RLock):
        self._lock = lock
        self.fault_hook = None  # see FakeGlobalAccelerator.fault_hook
        self._lbs: Dict[str, t.LoadBalancer] = {}  # arn -> LB
        self._regions: Dict[str, str] = {}  # arn -> region
        self._by_name: Dict[Tuple[str, str], str] = {}  # (region, name) -> arn

    # -- test/bench seeding helpers ---------------------------------------
    def create_load_balancer(
        self,
        name: str,
        region: str = "us-east-1",
        lb_type: str = "network",
        scheme: str = "internet-facing",
        state: str = t.LB_STATE_ACTIVE,
        dns_name: Optional[str] = None,
    ) -> t.LoadBalancer:
        with self._lock:
            if (region, name) in self._by_name:
                raise awserr.AWSAPIError(
                    f"A load balancer with the name '{name}' already exists",
                    "DuplicateLoadBalancerName",
                )
            kind = "net" if lb_type == "network" else "app"
            arn = (
                f"arn:aws:elasticloadbalancing:{region}:{_ACCOUNT}:"
                f"loadbalancer/{kind}/{name}/{uuid.uuid4().hex[:16]}"
            )
            if dns_name is None:
                suffix = _short()
                if lb_type == "network":
                    dns_name = f"{name}-{suffix}.elb.{region}.amazonaws.com"
                else:
                    prefix = "internal-" if scheme == "internal" else ""
                    dns_name = f"{prefix}{name}-{suffix}.{region}.elb.amazonaws.com"
            lb = t.LoadBalancer(
                load_balancer_arn=arn,
                load_balancer_name=name,
                dns_name=dns_name,
                state_code=state,
                type=lb_type,
                scheme=scheme,
            )
            self._lbs[arn] = lb
            self._regions[arn] = region
            self._by_name[(region, name)] = arn
            return _copy(lb)

    def set_state(self, name_or_arn: str, state: str):
        with self._lock:
            for arn, lb in self._lbs.items():
                if arn == name_or_arn or lb.load_balancer_name == name_or_arn:
                    lb.state_code = state
                    return
            raise awserr.LoadBalancerNotFoundException(name_or_arn)

    # -- API surface -------------------------------------------------------
    def describe_load_balancers(
        self,
        names: Optional[List[str]] = None,
        marker: Optional[str] = None,
        page_size: Optional[int] = None,
    ) -> Tuple[List[t.LoadBalancer], Optional[str]]:
        if self.fault_hook is not None:
            self.fault_hook("elbv2", "describe_load_balancers")
        with self._lock:
            lbs = sorted(self._lbs.values(), key=lambda x: x.load_balancer_arn)
            if names:
                found = [lb for lb in lbs if lb.load_balancer_name in names]
                missing = set(names) - {lb.load_balancer_name for lb in found}
                if missing:
                    # real ELBv2 fails the whole call for unknown names
                    raise awserr.LoadBalancerNotFoundException(
                        f"Load balancers '[{', '.join(sorted(missing))}]' not found"
                    )
                return [_copy(lb) for lb in found], None
            page, next_marker = _paginate(lbs, page_size, marker)
            return [_copy(lb) for lb in page], next_marker


class RegionalELBv2View:
    """Region-scoped view over the account-wide FakeELBv2 — real ELBv2
    clients are regional, so a us-east-1 client must not see eu-west-1
    load balancers (caught by the multi-region EndpointGroupBinding test)."""

    def __init__(self, elbv2: FakeELBv2, region: str):
        self._elbv2 = elbv2
        self.region = region

    def describe_load_balancers(self, names=None, marker=None, page_size=None):
        if self._elbv2.fault_hook is not None:
            self._elbv2.fault_hook("elbv2", "describe_load_balancers")
        with self._elbv2._lock:
            if names:
                found, missing = [], []
                for name in names:
                    arn = self._elbv2._by_name.get((self.region, name))
                    if arn is None:
                        missing.append(name)
                    else:
                        found.append(self._elbv2._lbs[arn])
                if missing:
                    raise awserr.LoadBalancerNotFoundException(
                        f"Load balancers '[{', '.join(sorted(missing))}]' not found"
                    )
                return [_copy(lb) for lb in found], None
            lbs = sorted(
                (
                    lb
                    for arn, lb in self._elbv2._lbs.items()
                    if self._elbv2._regions.get(arn) == self.region
                ),
                key=lambda x: x.load_balancer_arn,
            )
            page, next_marker = _paginate(lbs, page_size, marker)
            return [_copy(lb) for lb in page], next_marker


class FakeGlobalAccelerator:
    def __init__(self, lock: threading.RLock, deploy_after_describes: int = 1):
        self._lock = lock
        self.deploy_after_describes = deploy_after_describes
        self.call_counts: Dict[str, int] = {}
        # fault injection (chaos tests): callable (service, op) that may
        # raise an AWSAPIError to simulate throttles/5xx; fires at the top
        # of every operation, before any state mutation
        self.fault_hook = None
        self._accelerators: Dict[str, t.Accelerator] = {}
        self._pending: Dict[str, int] = {}  # arn -> remaining IN_PROGRESS describes
        self._tags: Dict[str, Dict[str, str]] = {}
        self._listeners: Dict[str, t.Listener] = {}  # listener arn -> listener
        self._listener_owner: Dict[str, str] = {}  # listener arn -> accelerator arn
        self._listeners_by_acc: Dict[str, List[str]] = {}  # acc arn -> [listener arn]
        self._endpoint_groups: Dict[str, t.EndpointGroup] = {}
        self._eg_owner: Dict[str, str] = {}  # endpoint group arn -> listener arn
        self._egs_by_listener: Dict[str, List[str]] = {}  # listener arn -> [eg arn]

    def _count(self, op: str):
        self.call_counts[op] = self.call_counts.get(op, 0) + 1
        if self.fault_hook is not None:
            self.fault_hook("globalaccelerator", op)

    # -- lifecycle helper --------------------------------------------------
    def _mutated(self, arn: str):
        self._accelerators[arn].status = t.ACCELERATOR_STATUS_IN_PROGRESS
        self._pending[arn] = self.deploy_after_describes

    # -- accelerators ------------------------------------------------------
    def create_accelerator(
        self,
        name: str,
        ip_address_type: str = t.IP_ADDRESS_TYPE_DUAL_STACK,
        enabled: bool = True,
        tags: Optional[List[t.Tag]] = None,
    ) -> t.Accelerator:
        with self._lock:
            self._count('create_accelerator')
            arn = f"arn:aws:globalaccelerator::{_ACCOUNT}:accelerator/{uuid.uuid4()}"
            acc = t.Accelerator(
                accelerator_arn=arn,
                name=name,
                dns_name=f"a{uuid.uuid4().hex[:13]}.awsglobalaccelerator.com",
                enabled=enabled,
                status=t.ACCELERATOR_STATUS_IN_PROGRESS,
                ip_address_type=ip_address_type,
            )
            self._accelerators[arn] = acc
            self._pending[arn] = self.deploy_after_describes
            self._tags[arn] = {tag.key: tag.value for tag in (tags or [])}
            return _copy(acc)

    def describe_accelerator(self, arn: str) -> t.Accelerator:
        with self._lock:
            self._count('describe_accelerator')
            acc = self._accelerators.get(arn)
            if acc is None:
                raise awserr.AcceleratorNotFoundException(arn)
            remaining = self._pending.get(arn, 0)
            if remaining > 0:
                self._pending[arn] = remaining - 1
            else:
                acc.status = t.ACCELERATOR_STATUS_DEPLOYED
            return _copy(acc)

    def list_accelerators(self, max_results: Optional[int] = None, next_token=None):
        with self._lock:
            self._count('list_accelerators')
            items = sorted(self._accelerators.values(), key=lambda a: a.accelerator_arn)
            page, token = _paginate(items, max_results, next_token)
            return [_copy(a) for a in page], token

    def update_accelerator(
        self,
        arn: str,
        name: Optional[str] = None,
        enabled: Optional[bool] = None,
        ip_address_type: Optional[str] = None,
    ) -> t.Accelerator:
        with self._lock:
            self._count('update_accelerator')
            acc = self._accelerators.get(arn)
            if acc is None:
                raise awserr.AcceleratorNotFoundException(arn)
            if name is not None:
                acc.name = name
            if enabled is not None:
                acc.enabled = enabled
            if ip_address_type is not None:
                acc.ip_address_type = ip_address_type
            self._mutated(arn)
            return _copy(acc)

    def delete_accelerator(self, arn: str):
        with self._lock:
            self._count('delete_accelerator')
            acc = self._accelerators.get(arn)
            if acc is None:
                raise awserr.AcceleratorNotFoundException(arn)
            if acc.enabled or acc.status != t.ACCELERATOR_STATUS_DEPLOYED:
                raise awserr.AcceleratorNotDisabledException(
                    f"accelerator {arn} must be disabled and deployed before deletion"
                )
            if self._listeners_by_acc.get(arn):
                raise awserr.AWSAPIError(
                    f"accelerator {arn} still has listeners", "AssociatedListenerFoundException"
                )
            del self._accelerators[arn]
            self._pending.pop(arn, None)
            self._tags.pop(arn, None)

    def list_tags_for_resource(self, arn: str) -> List[t.Tag]:
        with self._lock:
            self._count('list_tags_for_resource')
            if arn not in self._tags:
                raise awserr.AcceleratorNotFoundException(arn)
            return [t.Tag(k, v) for k, v in self._tags[arn].items()]

    def tag_resource(self, arn: str, tags: List[t.Tag]):
        with self._lock:
            self._count('tag_resource')
            if arn not in self._tags:
                raise awserr.AcceleratorNotFoundException(arn)
            self._tags[arn].update({tag.key: tag.value for tag in tags})

    # -- listeners ---------------------------------------------------------
    def create_listener(
        self,
        accelerator_arn: str,
        port_ranges: List[t.PortRange],
        protocol: str,
        client_affinity: str = t.CLIENT_AFFINITY_NONE,
    ) -> t.Listener:
        with self._lock:
            self._count('create_listener')
            if accelerator_arn not in self._accelerators:
                raise awserr.AcceleratorNotFoundException(accelerator_arn)
            arn = f"{accelerator_arn}/listener/{_short()}"
            listener = t.Listener(
                listener_arn=arn,
                port_ranges=[_copy(p) for p in port_ranges],
                protocol=protocol,
                client_affinity=client_affinity,
            )
            self._listeners[arn] = listener
            self._listener_owner[arn] = accelerator_arn
            self._listeners_by_acc.setdefault(accelerator_arn, []).append(arn)
            self._mutated(accelerator_arn)
            return _copy(listener)

    def list_listeners(self, accelerator_arn: str, max_results=None, next_token=None):
        with self._lock:
            self._count('list_listeners')
            if accelerator_arn not in self._accelerators:
                raise awserr.AcceleratorNotFoundException(accelerator_arn)
            items = sorted(self._listeners_by_acc.get(accelerator_arn, []))
            page, token = _paginate(items, max_results, next_token)
            return [_copy(self._listeners[l]) for l in page], token

    def update_listener(
        self,
        listener_arn: str,
        port_ranges: Optional[List[t.PortRange]] = None,
        protocol: Optional[str] = None,
        client_affinity: Optional[str] = None,
    ) -> t.Listener:
        with self._lock:
            self._count('update_listener')
            listener = self._listeners.get(listener_arn)
            if listener is None:
                raise awserr.ListenerNotFoundException(listener_arn)
            if port_ranges is not None:
                listener.port_ranges = [_copy(p) for p in port_ranges]
            if protocol is not None:
                listener.protocol = protocol
            if client_affinity is not None:
                listener.client_affinity = client_affinity
            self._mutated(self._listener_owner[listener_arn])
            return _copy(listener)

    def delete_listener(self, listener_arn: str):
        with self._lock:
            self._count('delete_listener')
            if listener_arn not in self._listeners:
                raise awserr.ListenerNotFoundException(listener_arn)
            if self._egs_by_listener.get(listener_arn):
                raise awserr.AWSAPIError(
                    f"listener {listener_arn} still has endpoint groups",
                    "AssociatedEndpointGroupFoundException",
                )
            acc_arn = self._listener_owner.pop(listener_arn)
            del self._listeners[listener_arn]
            owned_list = self._listeners_by_acc.get(acc_arn)
            if owned_list and listener_arn in owned_list:
                owned_list.remove(listener_arn)
            if acc_arn in self._accelerators:
                self._mutated(acc_arn)

    # -- endpoint groups ---------------------------------------------------
    def create_endpoint_group(
        self,
        listener_arn: str,
        endpoint_group_region: str,
        endpoint_configurations: Optional[List[t.EndpointConfiguration]] = None,
    ) -> t.EndpointGroup:
        with self._lock:
            self._count('create_endpoint_group')
            if listener_arn not in self._listeners:
                raise awserr.ListenerNotFoundException(listener_arn)
            arn = f"{listener_arn}/endpoint-group/{_short()}"
            eg = t.EndpointGroup(
                endpoint_group_arn=arn,
                endpoint_group_region=endpoint_group_region,
                endpoint_descriptions=[
                    _to_description(c) for c in (endpoint_configurations or [])
                ],
            )
            self._endpoint_groups[arn] = eg
            self._eg_owner[arn] = listener_arn
            self._egs_by_listener.setdefault(listener_arn, []).append(arn)
            self._mutated(self._listener_owner[listener_arn])
            return _copy(eg)

    def list_endpoint_groups(self, listener_arn: str, max_results=None, next_token=None):
        with self._lock:
            self._count('list_endpoint_groups')
            if listener_arn not in self._listeners:
                raise awserr.ListenerNotFoundException(listener_arn)
            items = sorted(self._egs_by_listener.get(listener_arn, []))
            page, token = _paginate(items, max_results, next_token)
            return [_copy(self._endpoint_groups[e]) for e in page], token

    def describe_endpoint_group(self, endpoint_group_arn: str) -> t.EndpointGroup:
        with self._lock:
            self._count('describe_endpoint_group')
            eg = self._endpoint_groups.get(endpoint_group_arn)
            if eg is None:
                raise awserr.EndpointGroupNotFoundException(endpoint_group_arn)
            return _copy(eg)

    def update_endpoint_group(
        self,
        endpoint_group_arn: str,
        endpoint_configurations: Optional[List[t.EndpointConfiguration]] = None,
    ) -> t.EndpointGroup:
        """Real-AWS semantics: EndpointConfigurations REPLACES the whole set."""
        with self._lock:
            self._count('update_endpoint_group')
            eg = self._endpoint_groups.get(endpoint_group_arn)
            if eg is None:
                raise awserr.EndpointGroupNotFoundException(endpoint_group_arn)
            if endpoint_configurations is not None:
                eg.endpoint_descriptions = [
                    _to_description(c) for c in endpoint_configurations
                ]
            self._touch_owner(endpoint_group_arn)
            return _copy(eg)

    def add_endpoints(
        self,
        endpoint_group_arn: str,
        endpoint_configurations: List[t.EndpointConfiguration],
    ) -> List[t.EndpointDescription]:
        with self._lock:
            self._count('add_endpoints')
            eg = self._endpoint_groups.get(endpoint_group_arn)
            if eg is None:
                raise awserr.EndpointGroupNotFoundException(endpoint_group_arn)
            added = []
            for config in endpoint_configurations:
                existing = next(
                    (
                        d
                        for d in eg.endpoint_descriptions
                        if d.endpoint_id == config.endpoint_id
                    ),
                    None,
                )
                if existing is None:
                    existing = _to_description(config)
                    eg.endpoint_descriptions.append(existing)
                else:
                    existing.weight = config.weight
                    existing.client_ip_preservation_enabled = (
                        config.client_ip_preservation_enabled
                    )
                added.append(_copy(existing))
            self._touch_owner(endpoint_group_arn)
            return added

    def remove_endpoints(self, endpoint_group_arn: str, endpoint_ids: List[str]):
        with self._lock:
            self._count('remove_endpoints')
            eg = self._endpoint_groups.get(endpoint_group_arn)
            if eg is None:
                raise awserr.EndpointGroupNotFoundException(endpoint_group_arn)
            eg.endpoint_descriptions = [
                d for d in eg.endpoint_descriptions if d.endpoint_id not in endpoint_ids
            ]
            self._touch_owner(endpoint_group_arn)

    def delete_endpoint_group(self, endpoint_group_arn: str):
        with self._lock:
            self._count('delete_endpoint_group')
            if endpoint_group_arn not in self._endpoint_groups:
                raise awserr.EndpointGroupNotFoundException(endpoint_group_arn)
            listener_arn = self._eg_owner.pop(endpoint_group_arn)
            del self._endpoint_groups[endpoint_group_arn]
            owned_list = self._egs_by_listener.get(listener_arn)
            if owned_list and endpoint_group_arn in owned_list:
                owned_list.remove(endpoint_group_arn)
            acc_arn = self._listener_owner.get(listener_arn)
            if acc_arn:
                self._mutated(acc_arn)

    def _touch_owner(self, endpoint_group_arn: str):
        listener_arn = self._eg_owner.get(endpoint_group_arn)
        acc_arn = self._listener_owner.get(listener_arn) if listener_arn else None
        if acc_arn:
            self._mutated(acc_arn)


def _normalize_record_name(name: str) -> str:
    """Route53 stores names dot-terminated with '*' escaped to '\\052'."""
    if not name.endswith("."):
        name += "."
    return name.replace("*", "\\052", 1)


class FakeRoute53:
    def __init__(self, lock: threading.RLock):
        self._lock = lock
        self.fault_hook = None  # see FakeGlobalAccelerator.fault_hook
        self._zones: Dict[str, t.HostedZone] = {}
        # zone id -> {(name, type) -> ResourceRecordSet}
        self._records: Dict[str, Dict[Tuple[str, str], t.ResourceRecordSet]] = {}

    # -- seeding helper ----------------------------------------------------
    def create_hosted_zone(self, name: str) -> t.HostedZone:
        with self._lock:
            if not name.endswith("."):
                name += "."
            zone = t.HostedZone(id=f"Z{uuid.uuid4().hex[:13].upper()}", name=name)
            self._zones[zone.id] = zone
            self._records[zone.id] = {}
            return _copy(zone)

    # -- API surface -------------------------------------------------------
    def list_hosted_zones(self, max_items: Optional[int] = None, marker=None):
        if self.fault_hook is not None:
            self.fault_hook("route53", "list_hosted_zones")
        with self._lock:
            zones = sorted(self._zones.values(), key=lambda z: z.name)
            page, token = _paginate(zones, max_items, marker)
            return [_copy(z) for z in page], token

    def list_hosted_zones_by_name(self, dns_name: str, max_items: Optional[int] = None):
        """Zones with name lexicographically >= dns_name, like the real API."""
        if self.fault_hook is not None:
            self.fault_hook("route53", "list_hosted_zones_by_name")
        with self._lock:
            zones = sorted(self._zones.values(), key=lambda z: z.name)
            after = [z for z in zones if z.name >= dns_name]
            page, _ = _paginate(after, max_items, None)
            return [_copy(z) for z in page]

    def list_resource_record_sets(
        self, zone_id: str, max_items=None, start_token=None, start_record_name=None
    ):
        """start_record_name mirrors the real API's StartRecordName: begin
        at the first record whose name >= the (normalized) given name."""
        if self.fault_hook is not None:
            self.fault_hook("route53", "list_resource_record_sets")
        with self._lock:
            records = self._records.get(zone_id)
            if records is None:
                raise awserr.NoSuchHostedZone(zone_id)
            items = [records[k] for k in sorted(records)]
            if start_record_name is not None:
                start = _normalize_record_name(start_record_name)
                items = [r for r in items if r.name >= start]
            page, token = _paginate(items, max_items, start_token)
            return [_copy(r) for r in page], token

    def change_resource_record_sets(self, zone_id: str, changes: List[t.Change]):
        if self.fault_hook is not None:
            self.fault_hook("route53", "change_resource_record_sets")
        with self._lock:
            records = self._records.get(zone_id)
            if records is None:
                raise awserr.NoSuchHostedZone(zone_id)
            # validate the whole batch first (changes are atomic in Route53)
            staged = dict(records)
            for change in changes:
                rs = _copy(change.record_set)
                rs.name = _normalize_record_name(rs.name)
                if rs.alias_target is not None and not rs.alias_target.dns_name.endswith("."):
                    # Route53 stores alias DNS names dot-terminated
                    rs.alias_target.dns_name += "."
                key = (rs.name, rs.type)
                if change.action == t.CHANGE_ACTION_CREATE:
                    if key in staged:
                        raise awserr.InvalidChangeBatch(
                            f"record {key} already exists"
                        )
                    staged[key] = rs
                elif change.action == t.CHANGE_ACTION_UPSERT:
                    staged[key] = rs
                elif change.action == t.CHANGE_ACTION_DELETE:
                    if key not in staged:
                        raise awserr.InvalidChangeBatch(f"record {key} not found")
                    del staged[key]
                else:
                    raise awserr.InvalidChangeBatch(
                        f"unknown action {change.action}"
                    )
            self._records[zone_id] = staged


class FakeAWSBackend:
    """The full in-memory AWS, shared by every regional client in tests."""

    def __init__(self, deploy_after_describes: int = 1):
        lock = threading.RLock()
        self.lock = lock
        self.elbv2 = FakeELBv2(lock)
        self.ga = FakeGlobalAccelerator(lock, deploy_after_describes)
        self.route53 = FakeRoute53(lock)

    def set_fault_hook(self, hook):
        """Install ``hook(service, op)`` on every service; it may raise an
        AWSAPIError to simulate throttling/5xx faults.  Fires before any
        state mutation, so a failed call never half-applies.  None clears."""
        self.elbv2.fault_hook = hook
        self.ga.fault_hook = hook
        self.route53.fault_hook = hook


def _copy(obj):
    from ...apis.meta import deep_copy

    return deep_copy(obj)


def _to_description(config: t.EndpointConfiguration) -> t.EndpointDescription:
    return t.EndpointDescription(
        endpoint_id=config.endpoint_id,
        weight=config.weight,
        client_ip_preservation_enabled=config.client_ip_preservation_enabled,
    )
