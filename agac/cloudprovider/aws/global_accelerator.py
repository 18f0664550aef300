"""Global Accelerator resource manager.

Behavior parity with reference ``pkg/cloudprovider/aws/global_accelerator.go``
(the largest component, ~1000 LoC there): idempotent ensure/update/cleanup of
the accelerator → listener → endpoint-group triple, tag-based ownership,
listener derivation from Service/Ingress specs, and endpoint management for
the EndpointGroupBinding CRD.

Deliberate fixes over the reference (documented per method):
- the ingress create path no longer swallows listener-create errors
  (reference ``global_accelerator.go:241-244`` returns nil error there);
- ``update_endpoint_weight`` preserves the other endpoints in the group
  (the reference sends UpdateEndpointGroup with a single configuration,
  which on real AWS replaces the entire endpoint set).
"""

from __future__ import annotations

import json
import logging
from typing import List, Optional, Tuple

from ... import metrics
from ...apis import (
    ALB_LISTEN_PORTS_ANNOTATION,
    AWS_GLOBAL_ACCELERATOR_IP_ADDRESS_TYPE_ANNOTATION,
    AWS_GLOBAL_ACCELERATOR_NAME_ANNOTATION,
    AWS_GLOBAL_ACCELERATOR_TAGS_ANNOTATION,
    CLIENT_IP_PRESERVATION_ANNOTATION,
)
from . import errors as awserr
from . import types as t

logger = logging.getLogger(__name__)

# Ownership tag schema — byte-for-byte parity with the reference
# (global_accelerator.go:24-28) so both controllers can co-manage resources.
GLOBAL_ACCELERATOR_MANAGED_TAG_KEY = "aws-global-accelerator-controller-managed"
GLOBAL_ACCELERATOR_OWNER_TAG_KEY = "aws-global-accelerator-owner"
GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY = "aws-global-accelerator-target-hostname"
GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY = "aws-global-accelerator-cluster"

DEFAULT_LB_NOT_ACTIVE_RETRY = 30.0  # seconds (reference :127)


# ---------------------------------------------------------------------------
# Pure helpers (unit-testable; the reference's test tables cover these)
# ---------------------------------------------------------------------------
def accelerator_owner_tag_value(resource: str, ns: str, name: str) -> str:
    return f"{resource}/{ns}/{name}"


def accelerator_tags(obj) -> List[t.Tag]:
    """Parse the global-accelerator-tags annotation 'k=v,k2=v2' into tags;
    malformed entries are skipped (reference :35-51)."""
    results = []
    raw = obj.metadata.annotations.get(AWS_GLOBAL_ACCELERATOR_TAGS_ANNOTATION, "")
    for pair in raw.split(","):
        parts = pair.split("=")
        if len(parts) != 2:
            continue
        results.append(t.Tag(key=parts[0], value=parts[1]))
    return results


def accelerator_name(resource: str, obj) -> str:
    name = obj.metadata.annotations.get(AWS_GLOBAL_ACCELERATOR_NAME_ANNOTATION, "")
    if name:
        return name
    return f"{resource}-{obj.metadata.namespace}-{obj.metadata.name}"


def resolve_ip_address_type(annotation_value: str) -> str:
    """ip-address-type annotation → IPV4 / DUAL_STACK, defaulting to
    DUAL_STACK for empty or unknown values (reference :676-687)."""
    if annotation_value in ("ipv4", "IPV4"):
        return t.IP_ADDRESS_TYPE_IPV4
    if annotation_value in ("dualstack", "DUAL_STACK", ""):
        return t.IP_ADDRESS_TYPE_DUAL_STACK
    logger.warning("Unknown IP address type %s, defaulting to DUAL_STACK", annotation_value)
    return t.IP_ADDRESS_TYPE_DUAL_STACK


def tags_contains_all_values(tags: List[t.Tag], target: dict) -> bool:
    actual = {tag.key: tag.value for tag in tags}
    return all(actual.get(k) == v for k, v in target.items())


def listener_for_service(svc) -> Tuple[List[int], str]:
    """All service ports; protocol is UDP iff the LAST explicitly-typed port
    says so (reference :503-515 keeps the last seen protocol)."""
    ports = []
    protocol = t.PROTOCOL_TCP
    for p in svc.spec.ports:
        ports.append(p.port)
        if p.protocol.lower() == "udp":
            protocol = t.PROTOCOL_UDP
        elif p.protocol.lower() == "tcp":
            protocol = t.PROTOCOL_TCP
    return ports, protocol


def listener_for_ingress(ingress) -> Tuple[List[int], str]:
    """Ports from the alb.ingress.kubernetes.io/listen-ports JSON annotation
    when present, else default-backend + rule backend service ports; always
    TCP (ALB has no UDP) (reference :522-557)."""
    ports: List[int] = []
    protocol = t.PROTOCOL_TCP
    raw = ingress.metadata.annotations.get(ALB_LISTEN_PORTS_ANNOTATION)
    if raw is not None:
        try:
            entries = json.loads(raw)
        except (ValueError, TypeError) as e:
            logger.error("invalid listen-ports annotation %r: %s", raw, e)
            return ports, protocol
        if not isinstance(entries, list):
            logger.error("invalid listen-ports annotation %r: not a list", raw)
            return ports, protocol
        for entry in entries:
            if not isinstance(entry, dict):
                continue
            if entry.get("HTTP"):
                ports.append(int(entry["HTTP"]))
            if entry.get("HTTPS"):
                ports.append(int(entry["HTTPS"]))
        return ports, protocol

    if ingress.spec.default_backend is not None and ingress.spec.default_backend.service is not None:
        ports.append(ingress.spec.default_backend.service.port.number)
    for rule in ingress.spec.rules:
        if rule.http is not None:
            for path in rule.http.paths:
                if path.backend.service is not None:
                    ports.append(path.backend.service.port.number)
    return ports, protocol


def listener_protocol_changed_from_service(listener: t.Listener, svc) -> bool:
    _, protocol = listener_for_service(svc)
    return listener.protocol != protocol


def listener_protocol_changed_from_ingress(listener: t.Listener, ingress) -> bool:
    # ALB is HTTP/TCP only, so a GA listener for an Ingress must be TCP.
    return listener.protocol != t.PROTOCOL_TCP


def _port_sets_differ(listener_ports: List[int], desired_ports: List[int]) -> bool:
    """The reference's port-count trick (:458-474): any port seen exactly
    once across (listener ∪ desired) means drift."""
    count: dict = {}
    for p in listener_ports:
        count[p] = count.get(p, 0) + 1
    for p in desired_ports:
        count[p] = count.get(p, 0) + 1
    return any(v <= 1 for v in count.values())


def listener_port_changed_from_service(listener: t.Listener, svc) -> bool:
    ports, _ = listener_for_service(svc)
    return _port_sets_differ([pr.from_port for pr in listener.port_ranges], ports)


def listener_port_changed_from_ingress(listener: t.Listener, ingress) -> bool:
    ports, _ = listener_for_ingress(ingress)
    return _port_sets_differ([pr.from_port for pr in listener.port_ranges], ports)


def endpoint_contains_lb(endpoint_group: t.EndpointGroup, lb) -> bool:
    return any(
        d.endpoint_id == lb.load_balancer_arn
        for d in endpoint_group.endpoint_descriptions
    )


# ---------------------------------------------------------------------------
# Resource manager
# ---------------------------------------------------------------------------
class GlobalAcceleratorMixin:
    """Methods bound into ``agac.cloudprovider.aws.client.AWS``."""

    # -- listing by ownership tags (reference :62-110) --------------------
    def list_global_accelerator_by_hostname(self, hostname: str, cluster_name: str):
        return self._list_accelerators_matching(
            {
                GLOBAL_ACCELERATOR_MANAGED_TAG_KEY: "true",
                GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY: hostname,
                GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY: cluster_name,
            }
        )

    def list_global_accelerator_by_resource(
        self, cluster_name: str, resource: str, ns: str, name: str
    ):
        return self._list_accelerators_matching(
            {
                GLOBAL_ACCELERATOR_MANAGED_TAG_KEY: "true",
                GLOBAL_ACCELERATOR_OWNER_TAG_KEY: accelerator_owner_tag_value(
                    resource, ns, name
                ),
                GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY: cluster_name,
            }
        )

    def _list_accelerators_matching(self, target_tags: dict):
        result = []
        for accelerator in self._list_accelerators():
            tags = self._list_tags_for_accelerator(accelerator.accelerator_arn)
            if tags_contains_all_values(tags, target_tags):
                result.append(accelerator)
            else:
                logger.debug(
                    "Global Accelerator %s does not have match tags",
                    accelerator.accelerator_arn,
                )
        return result

    # -- ensure (reference :112-211) ---------------------------------------
    def ensure_global_accelerator_for_service(
        self, svc, lb_ingress, cluster_name: str, lb_name: str, region: str,
        hint_arn: Optional[str] = None,
    ):
        """Returns (accelerator_arn | None, created, retry_after_seconds).

        ``hint_arn`` short-circuits the O(#accelerators × ListTags) discovery
        scan (reference SURVEY §3.2 hot loop): the hinted accelerator is
        fetched directly and its ownership tags VERIFIED before use; any
        mismatch or error falls back to the full scan, so behavior is
        identical — only steady-state API cost drops to O(1)."""
        return self._ensure_global_accelerator(
            obj=svc,
            resource="service",
            hostname=lb_ingress.hostname,
            cluster_name=cluster_name,
            lb_name=lb_name,
            region=region,
            listener_spec=listener_for_service,
            protocol_changed=listener_protocol_changed_from_service,
            port_changed=listener_port_changed_from_service,
            hint_arn=hint_arn,
        )

    def ensure_global_accelerator_for_ingress(
        self, ingress, lb_ingress, cluster_name: str, lb_name: str, region: str,
        hint_arn: Optional[str] = None,
    ):
        """Same as the service path, with ingress listener derivation.
        Unlike the reference (:241-244) a listener-create failure here is an
        error (and triggers the partial-create cleanup)."""
        return self._ensure_global_accelerator(
            obj=ingress,
            resource="ingress",
            hostname=lb_ingress.hostname,
            cluster_name=cluster_name,
            lb_name=lb_name,
            region=region,
            listener_spec=listener_for_ingress,
            protocol_changed=listener_protocol_changed_from_ingress,
            port_changed=listener_port_changed_from_ingress,
            hint_arn=hint_arn,
        )

    def _verified_hint(
        self,
        hint_arn: str,
        cluster_name: str,
        resource: str,
        ns: str,
        name: str,
        hostname: str,
    ):
        """Fetch the hinted accelerator and verify our ownership tags —
        including the target-hostname tag, so a resource with multiple LB
        hostnames (one owned accelerator per hostname) never resolves the
        hint to the *other* hostname's accelerator; any mismatch falls back
        to the full ``list_global_accelerator_by_resource`` scan, which is
        the reference's only discovery path (global_accelerator.go:87-110).
        Returns [accelerator] or None to force the full scan."""
        try:
            accelerator = self._get_accelerator(hint_arn)
            tags = self._list_tags_for_accelerator(hint_arn)
        except Exception:
            metrics.observe_hint("globalaccelerator", "error")
            return None
        if tags_contains_all_values(
            tags,
            {
                GLOBAL_ACCELERATOR_MANAGED_TAG_KEY: "true",
                GLOBAL_ACCELERATOR_OWNER_TAG_KEY: accelerator_owner_tag_value(
                    resource, ns, name
                ),
                GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY: cluster_name,
                GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY: hostname,
            },
        ):
            metrics.observe_hint("globalaccelerator", "hit")
            # stash for same-reconcile reuse by _accelerator_changed (the
            # reference re-lists tags it fetched moments earlier in the
            # same sync, global_accelerator.go:87-110 then :412-437; the
            # data is identical modulo a millisecond write race)
            self._hint_verified_tags = (accelerator.accelerator_arn, tags)
            return [accelerator]
        metrics.observe_hint("globalaccelerator", "stale")
        return None

    def _ensure_global_accelerator(
        self,
        obj,
        resource: str,
        hostname: str,
        cluster_name: str,
        lb_name: str,
        region: str,
        listener_spec,
        protocol_changed,
        port_changed,
        hint_arn: Optional[str] = None,
    ):
        # one-reconcile scope for the hint-tag reuse (the controller builds
        # a fresh AWS per reconcile like the reference's NewAWS, but guard
        # direct library reuse of one instance across ensures)
        self._hint_verified_tags = None
        lb = self.get_load_balancer(lb_name)
        if lb.dns_name != hostname:
            raise ValueError(f"LoadBalancer's DNS name is not matched: {lb.dns_name}")
        if lb.state_code != t.LB_STATE_ACTIVE:
            logger.warning(
                "LoadBalancer %s is not Active: %s", lb.load_balancer_arn, lb.state_code
            )
            return None, False, self.lb_not_active_retry

        logger.info("LoadBalancer is %s", lb.load_balancer_arn)
        accelerators = None
        if hint_arn:
            accelerators = self._verified_hint(
                hint_arn,
                cluster_name,
                resource,
                obj.metadata.namespace,
                obj.metadata.name,
                hostname,
            )
        if accelerators is None:
            accelerators = self.list_global_accelerator_by_resource(
                cluster_name, resource, obj.metadata.namespace, obj.metadata.name
            )
        if not accelerators:
            logger.info("Creating Global Accelerator for %s", lb.dns_name)
            created_arn = self._create_global_accelerator(
                lb, obj, resource, cluster_name, region, listener_spec
            )
            return created_arn, True, 0.0

        for accelerator in accelerators:
            logger.info(
                "Updating existing Global Accelerator %s", accelerator.accelerator_arn
            )
            self._update_global_accelerator(
                accelerator,
                lb,
                obj,
                resource,
                region,
                listener_spec,
                protocol_changed,
                port_changed,
            )
        return accelerators[0].accelerator_arn, False, 0.0

    def _create_global_accelerator(
        self, lb, obj, resource: str, cluster_name: str, region: str, listener_spec
    ) -> str:
        """Create the accelerator→listener→endpoint-group triple; on partial
        failure, clean up what was created then re-raise
        (reference :142-147, :213-252)."""
        ip_address_type = obj.metadata.annotations.get(
            AWS_GLOBAL_ACCELERATOR_IP_ADDRESS_TYPE_ANNOTATION, ""
        )
        accelerator = self._create_accelerator(
            accelerator_name(resource, obj),
            cluster_name,
            accelerator_owner_tag_value(
                resource, obj.metadata.namespace, obj.metadata.name
            ),
            lb.dns_name,
            ip_address_type,
            accelerator_tags(obj),
        )
        try:
            ports, protocol = listener_spec(obj)
            listener = self._create_listener(accelerator, ports, protocol)
            ip_preserve = (
                obj.metadata.annotations.get(CLIENT_IP_PRESERVATION_ANNOTATION) == "true"
            )
            self._create_endpoint_group(
                listener, lb.load_balancer_arn, region, ip_preserve
            )
        except Exception:
            logger.warning(
                "Failed to create Global Accelerator, but some resources are "
                "created, so cleanup %s",
                accelerator.accelerator_arn,
            )
            try:
                self.cleanup_global_accelerator(accelerator.accelerator_arn)
            except Exception:
                logger.exception(
                    "cleanup of partially-created accelerator %s failed",
                    accelerator.accelerator_arn,
                )
            raise
        return accelerator.accelerator_arn

    # -- update / drift repair (reference :290-410) ------------------------
    def _update_global_accelerator(
        self,
        accelerator,
        lb,
        obj,
        resource: str,
        region: str,
        listener_spec,
        protocol_changed,
        port_changed,
    ):
        if self._accelerator_changed(accelerator, lb.dns_name, resource, obj):
            self._update_accelerator(
                accelerator.accelerator_arn,
                accelerator_name(resource, obj),
                accelerator_owner_tag_value(
                    resource, obj.metadata.namespace, obj.metadata.name
                ),
                lb.dns_name,
                accelerator_tags(obj),
            )

        try:
            listener = self.get_listener(accelerator.accelerator_arn)
        except awserr.ListenerNotFoundException:
            ports, protocol = listener_spec(obj)
            listener = self._create_listener(accelerator, ports, protocol)
        if protocol_changed(listener, obj) or port_changed(listener, obj):
            logger.info("Listener is changed, so updating: %s", listener.listener_arn)
            ports, protocol = listener_spec(obj)
            listener = self._update_listener(listener, ports, protocol)

        ip_preserve = (
            obj.metadata.annotations.get(CLIENT_IP_PRESERVATION_ANNOTATION) == "true"
        )
        try:
            endpoint_group = self.get_endpoint_group(listener.listener_arn)
        except awserr.EndpointGroupNotFoundException:
            endpoint_group = self._create_endpoint_group(
                listener, lb.load_balancer_arn, region, ip_preserve
            )
        if not endpoint_contains_lb(endpoint_group, lb):
            logger.info(
                "Endpoint Group is changed, so updating: %s",
                endpoint_group.endpoint_group_arn,
            )
            self._update_endpoint_group(
                endpoint_group, lb.load_balancer_arn, ip_preserve
            )
        logger.info("All resources are synced: %s", accelerator.accelerator_arn)

    def _accelerator_changed(self, accelerator, hostname, resource, obj) -> bool:
        """Drift predicate (reference :412-437): disabled, renamed, or
        missing/stale ownership+user tags.  A tag-listing failure is logged
        and treated as no-drift, like the reference."""
        if not accelerator.enabled:
            return True
        if accelerator.name != accelerator_name(resource, obj):
            return True
        cached = getattr(self, "_hint_verified_tags", None)
        if cached is not None and cached[0] == accelerator.accelerator_arn:
            tags = cached[1]
        else:
            try:
                tags = self._list_tags_for_accelerator(accelerator.accelerator_arn)
            except Exception as e:
                logger.warning("listing tags for %s failed: %s", accelerator.accelerator_arn, e)
                return False
        target = {
            GLOBAL_ACCELERATOR_MANAGED_TAG_KEY: "true",
            GLOBAL_ACCELERATOR_OWNER_TAG_KEY: accelerator_owner_tag_value(
                resource, obj.metadata.namespace, obj.metadata.name
            ),
            GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY: hostname,
        }
        for tag in accelerator_tags(obj):
            target[tag.key] = tag.value
        return not tags_contains_all_values(tags, target)

    # -- cleanup (reference :254-288) ---------------------------------------
    def cleanup_global_accelerator(self, arn: str):
        accelerator, listener, endpoint_group = self._list_related(arn)
        if endpoint_group is not None:
            self._delete_endpoint_group(endpoint_group.endpoint_group_arn)
        if listener is not None:
            self._delete_listener(listener.listener_arn)
        if accelerator is not None:
            self._delete_accelerator(accelerator.accelerator_arn)

    def _list_related(self, arn: str):
        """Deliberate fix (docs/PARITY.md §5c): the reference's
        listRelatedGlobalAccelerator treats ANY error as "already deleted"
        (global_accelerator.go:274-288 returns nil on err), so a single
        throttled DescribeAccelerator during teardown makes
        CleanupGlobalAccelerator report success and the accelerator leaks
        FOREVER (nothing re-enqueues a forgotten cleanup).  Only the typed
        not-found errors mean "gone" — anything else propagates so the
        engine's rate-limited retry runs the cleanup again."""
        try:
            accelerator = self._get_accelerator(arn)
        except awserr.AcceleratorNotFoundException:
            return None, None, None
        try:
            listener = self.get_listener(accelerator.accelerator_arn)
        except awserr.ListenerNotFoundException:
            return accelerator, None, None
        try:
            endpoint_group = self.get_endpoint_group(listener.listener_arn)
        except awserr.EndpointGroupNotFoundException:
            return accelerator, listener, None
        return accelerator, listener, endpoint_group

    # -- EndpointGroupBinding support (reference :572-608) ------------------
    def add_lb_to_endpoint_group(
        self,
        endpoint_group: t.EndpointGroup,
        lb_name: str,
        ip_preserve: bool,
        weight: Optional[int],
    ):
        """Returns (endpoint_id | None, retry_after_seconds)."""
        lb = self.get_load_balancer(lb_name)
        if lb.state_code != t.LB_STATE_ACTIVE:
            logger.warning(
                "LoadBalancer %s is not Active: %s", lb.load_balancer_arn, lb.state_code
            )
            return None, self.lb_not_active_retry
        metrics.observe_aws_call("globalaccelerator", "AddEndpoints")
        added = self.ga.add_endpoints(
            endpoint_group.endpoint_group_arn,
            [
                t.EndpointConfiguration(
                    endpoint_id=lb.load_balancer_arn,
                    client_ip_preservation_enabled=ip_preserve,
                    weight=weight,
                )
            ],
        )
        if not added:
            raise RuntimeError("No endpoint is added")
        logger.info("Endpoint is added: %s", added[0].endpoint_id)
        return added[0].endpoint_id, 0.0

    def remove_lb_from_endpoint_group(
        self, endpoint_group: t.EndpointGroup, endpoint_id: str
    ):
        # (reference name has a typo: RemoveLBFromEdnpointGroup)
        metrics.observe_aws_call("globalaccelerator", "RemoveEndpoints")
        self.ga.remove_endpoints(endpoint_group.endpoint_group_arn, [endpoint_id])
        logger.info("Endpoint is removed: %s", endpoint_id)

    def update_endpoint_weight(
        self, endpoint_group: t.EndpointGroup, endpoint_id: str, weight: Optional[int]
    ):
        """Set one endpoint's weight, preserving the rest of the group.

        The reference (:931-947) sends UpdateEndpointGroup with only the one
        configuration — which on real AWS replaces the entire endpoint set.
        Here the current set is fetched and re-sent with just the target
        weight changed."""
        metrics.observe_aws_call("globalaccelerator", "DescribeEndpointGroup")
        current = self.ga.describe_endpoint_group(endpoint_group.endpoint_group_arn)
        configs = [
            t.EndpointConfiguration(
                endpoint_id=d.endpoint_id,
                weight=weight if d.endpoint_id == endpoint_id else d.weight,
                client_ip_preservation_enabled=d.client_ip_preservation_enabled,
            )
            for d in current.endpoint_descriptions
        ]
        metrics.observe_aws_call("globalaccelerator", "UpdateEndpointGroup")
        self.ga.update_endpoint_group(
            endpoint_group.endpoint_group_arn, endpoint_configurations=configs
        )
        logger.info("Endpoint weight is updated: %s", endpoint_id)

    def sync_endpoint_weights(
        self,
        endpoint_group: t.EndpointGroup,
        endpoint_ids,
        weight: Optional[int],
    ):
        """Set the weight of every id in ``endpoint_ids`` in one
        describe + one update (the reference loops per-endpoint
        UpdateEndpointGroup calls, egb/reconcile.go:199-206 — K×2 API
        round-trips instead of 2)."""
        metrics.observe_aws_call("globalaccelerator", "DescribeEndpointGroup")
        current = self.ga.describe_endpoint_group(endpoint_group.endpoint_group_arn)
        target = set(endpoint_ids)
        configs = [
            t.EndpointConfiguration(
                endpoint_id=d.endpoint_id,
                weight=weight if d.endpoint_id in target else d.weight,
                client_ip_preservation_enabled=d.client_ip_preservation_enabled,
            )
            for d in current.endpoint_descriptions
        ]
        metrics.observe_aws_call("globalaccelerator", "UpdateEndpointGroup")
        self.ga.update_endpoint_group(
            endpoint_group.endpoint_group_arn, endpoint_configurations=configs
        )
        logger.info("Endpoint weights are synced for %d endpoints", len(target))

    def describe_endpoint_group(self, endpoint_group_arn: str) -> t.EndpointGroup:
        metrics.observe_aws_call("globalaccelerator", "DescribeEndpointGroup")
        return self.ga.describe_endpoint_group(endpoint_group_arn)

    # -- accelerator primitives (reference :613-784) -------------------------
    def _get_accelerator(self, arn: str) -> t.Accelerator:
        metrics.observe_aws_call("globalaccelerator", "DescribeAccelerator")
        return self.ga.describe_accelerator(arn)

    def _list_accelerators(self) -> List[t.Accelerator]:
        accelerators = []
        token = None
        while True:
            metrics.observe_aws_call("globalaccelerator", "ListAccelerators")
            page, token = self.ga.list_accelerators(max_results=100, next_token=token)
            accelerators.extend(page)
            if token is None:
                return accelerators

    def _list_tags_for_accelerator(self, arn: str) -> List[t.Tag]:
        metrics.observe_aws_call("globalaccelerator", "ListTagsForResource")
        return self.ga.list_tags_for_resource(arn)

    def _create_accelerator(
        self, name, cluster_name, owner, hostname, ip_address_type, specified_tags
    ) -> t.Accelerator:
        logger.info("Creating Global Accelerator %s", name)
        tags = [
            t.Tag(GLOBAL_ACCELERATOR_MANAGED_TAG_KEY, "true"),
            t.Tag(GLOBAL_ACCELERATOR_OWNER_TAG_KEY, owner),
            t.Tag(GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY, hostname),
            t.Tag(GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY, cluster_name),
        ] + list(specified_tags)
        metrics.observe_aws_call("globalaccelerator", "CreateAccelerator")
        accelerator = self.ga.create_accelerator(
            name=name,
            ip_address_type=resolve_ip_address_type(ip_address_type),
            enabled=True,
            tags=tags,
        )
        logger.info("Global Accelerator is created: %s", accelerator.accelerator_arn)
        return accelerator

    def _update_accelerator(
        self, arn: str, name: str, owner: str, hostname: str, specified_tags
    ) -> t.Accelerator:
        logger.info("Updating Global Accelerator %s", arn)
        metrics.observe_aws_call("globalaccelerator", "UpdateAccelerator")
        updated = self.ga.update_accelerator(arn, name=name, enabled=True)
        tags = [
            t.Tag(GLOBAL_ACCELERATOR_MANAGED_TAG_KEY, "true"),
            t.Tag(GLOBAL_ACCELERATOR_OWNER_TAG_KEY, owner),
            t.Tag(GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY, hostname),
        ] + list(specified_tags)
        metrics.observe_aws_call("globalaccelerator", "TagResource")
        self.ga.tag_resource(arn, tags)
        return updated

    def _delete_accelerator(self, arn: str):
        """Disable, poll until DEPLOYED, then delete (reference :743-784).
        Poll cadence comes from self.poll_interval/self.poll_timeout."""
        logger.info("Disabling Global Accelerator %s", arn)
        metrics.observe_aws_call("globalaccelerator", "UpdateAccelerator")
        self.ga.update_accelerator(arn, enabled=False)

        import time as _time

        deadline = _time.monotonic() + self.poll_timeout
        while True:
            accelerator = self._get_accelerator(arn)
            if accelerator.status == t.ACCELERATOR_STATUS_DEPLOYED:
                logger.info(
                    "Global Accelerator %s is %s",
                    accelerator.accelerator_arn,
                    accelerator.status,
                )
                break
            logger.info(
                "Global Accelerator %s is %s, so waiting",
                accelerator.accelerator_arn,
                accelerator.status,
            )
            if _time.monotonic() >= deadline:
                raise TimeoutError(
                    f"accelerator {arn} did not reach DEPLOYED within "
                    f"{self.poll_timeout}s"
                )
            self.sleep(self.poll_interval)

        metrics.observe_aws_call("globalaccelerator", "DeleteAccelerator")
        self.ga.delete_accelerator(arn)
        logger.info("Global Accelerator is deleted: %s", arn)

    # -- listener primitives (reference :789-869) ----------------------------
    def get_listener(self, accelerator_arn: str) -> t.Listener:
        """Exactly-one invariant: 0 ⇒ ListenerNotFoundException,
        >1 ⇒ error (reference :789-813)."""
        listeners = []
        token = None
        while True:
            metrics.observe_aws_call("globalaccelerator", "ListListeners")
            page, token = self.ga.list_listeners(
                accelerator_arn, max_results=100, next_token=token
            )
            listeners.extend(page)
            if token is None:
                break
        if not listeners:
            raise awserr.ListenerNotFoundException(accelerator_arn)
        if len(listeners) > 1:
            raise RuntimeError("Too many listeners")
        return listeners[0]

    def _create_listener(self, accelerator, ports: List[int], protocol: str) -> t.Listener:
        port_ranges = [t.PortRange(from_port=p, to_port=p) for p in ports]
        metrics.observe_aws_call("globalaccelerator", "CreateListener")
        listener = self.ga.create_listener(
            accelerator.accelerator_arn,
            port_ranges,
            protocol,
            client_affinity=t.CLIENT_AFFINITY_NONE,
        )
        logger.info("Listener is created: %s", listener.listener_arn)
        return listener

    def _update_listener(self, listener, ports: List[int], protocol: str) -> t.Listener:
        port_ranges = [t.PortRange(from_port=p, to_port=p) for p in ports]
        metrics.observe_aws_call("globalaccelerator", "UpdateListener")
        updated = self.ga.update_listener(
            listener.listener_arn,
            port_ranges=port_ranges,
            protocol=protocol,
            client_affinity=t.CLIENT_AFFINITY_NONE,
        )
        logger.info("Listener is updated: %s", updated.listener_arn)
        return updated

    def _delete_listener(self, arn: str):
        metrics.observe_aws_call("globalaccelerator", "DeleteListener")
        self.ga.delete_listener(arn)
        logger.info("Listener is deleted: %s", arn)

    # -- endpoint group primitives (reference :874-1013) ---------------------
    def get_endpoint_group(self, listener_arn: str) -> t.EndpointGroup:
        """Exactly-one invariant like get_listener (reference :885-907)."""
        groups = []
        token = None
        while True:
            metrics.observe_aws_call("globalaccelerator", "ListEndpointGroups")
            page, token = self.ga.list_endpoint_groups(
                listener_arn, max_results=100, next_token=token
            )
            groups.extend(page)
            if token is None:
                break
        if not groups:
            raise awserr.EndpointGroupNotFoundException(listener_arn)
        if len(groups) > 1:
            raise RuntimeError("Too many endpoint groups")
        return groups[0]

    def _create_endpoint_group(
        self, listener, lb_arn: str, region: str, ip_preserve: bool
    ) -> t.EndpointGroup:
        metrics.observe_aws_call("globalaccelerator", "CreateEndpointGroup")
        endpoint_group = self.ga.create_endpoint_group(
            listener.listener_arn,
            endpoint_group_region=region,
            endpoint_configurations=[
                t.EndpointConfiguration(
                    endpoint_id=lb_arn, client_ip_preservation_enabled=ip_preserve
                )
            ],
        )
        logger.info(
            "EndpointGroup is created: %s", endpoint_group.endpoint_group_arn
        )
        return endpoint_group

    def _update_endpoint_group(
        self, endpoint_group, lb_arn: str, ip_preserve: bool
    ) -> t.EndpointGroup:
        metrics.observe_aws_call("globalaccelerator", "UpdateEndpointGroup")
        updated = self.ga.update_endpoint_group(
            endpoint_group.endpoint_group_arn,
            endpoint_configurations=[
                t.EndpointConfiguration(
                    endpoint_id=lb_arn, client_ip_preservation_enabled=ip_preserve
                )
            ],
        )
        logger.info("EndpointGroup is updated: %s", updated.endpoint_group_arn)
        return updated

    def _delete_endpoint_group(self, arn: str):
        metrics.observe_aws_call("globalaccelerator", "DeleteEndpointGroup")
        self.ga.delete_endpoint_group(arn)
        logger.info("EndpointGroup is deleted: %s", arn)
