"""Route53 resource manager.

Behavior parity with reference ``pkg/cloudprovider/aws/route53.go``:
A-ALIAS records pointing at the Global Accelerator DNS name, paired with
TXT ownership records whose value identifies the managing cluster and the
owning k8s resource.  Hosted zones are found by walking up parent domains;
wildcard names round-trip through Route53's ``\\052`` octal escaping.
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

from ... import metrics
from . import errors as awserr
from . import types as t

logger = logging.getLogger(__name__)

DEFAULT_GA_MISSING_RETRY = 60.0  # seconds (reference route53.go:72-76)
TXT_TTL = 300  # seconds (reference :276)
# hostname->zone hint lifetime: after this the parent-domain walk re-runs,
# so zone-topology changes (new more-specific zone) are honored within one
# TTL instead of only on NoSuchHostedZone (same magnitude as the TXT TTL)
ZONE_HINT_TTL = 300.0


# ---------------------------------------------------------------------------
# Pure helpers (covered by the reference's unit table route53_test.go)
# ---------------------------------------------------------------------------
def route53_owner_value(cluster_name: str, resource: str, ns: str, name: str) -> str:
    """TXT ownership record value — byte-for-byte parity including the
    embedded quotes (reference ``Route53OwnerValue``, route53.go:18-20)."""
    return (
        '"heritage=aws-global-accelerator-controller,cluster='
        + cluster_name
        + ","
        + resource
        + "/"
        + ns
        + "/"
        + name
        + '"'
    )


def replace_wildcards(s: str) -> str:
    """Route53 returns '*' as octal '\\052'; undo the first occurrence."""
    return s.replace("\\052", "*", 1)


def parent_domain(hostname: str) -> str:
    """Strip the leftmost label ('a.b.c' → 'b.c', 'c' → '')."""
    return ".".join(hostname.split(".")[1:])


def find_a_record(
    records: List[t.ResourceRecordSet], hostname: str
) -> Optional[t.ResourceRecordSet]:
    for record in records:
        if record.type == t.RR_TYPE_A and replace_wildcards(record.name) == hostname + ".":
            return record
    return None


def need_records_update(record: t.ResourceRecordSet, accelerator) -> bool:
    """True when the A-alias target drifted from the accelerator DNS name
    (reference :373-381; the stored alias DNS name is dot-terminated)."""
    if record.alias_target is None:
        return True
    return record.alias_target.dns_name != accelerator.dns_name + "."


# ---------------------------------------------------------------------------
# Resource manager
# ---------------------------------------------------------------------------
class Route53Mixin:
    """Methods bound into ``agac.cloudprovider.aws.client.AWS``."""

    def ensure_route53_for_service(
        self, svc, lb_ingress, hostnames: List[str], cluster_name: str,
        hint_arn: Optional[str] = None,
        zone_hints: Optional[dict] = None,
    ) -> Tuple[bool, float]:
        """``hint_arn`` short-circuits the by-hostname accelerator scan with
        tag-verified fallback (same contract as
        ensure_global_accelerator_for_service's hint).  ``zone_hints`` is a
        caller-owned hostname→HostedZone cache that skips the parent-domain
        ListHostedZonesByName walk; a hinted zone that turns out to be gone
        (NoSuchHostedZone on any operation) is dropped and the walk re-runs,
        so behavior is identical to the reference's per-reconcile walk
        (route53.go:335-358)."""
        return self._ensure_route53(
            lb_ingress.hostname,
            hostnames,
            cluster_name,
            "service",
            svc.metadata.namespace,
            svc.metadata.name,
            hint_arn=hint_arn,
            zone_hints=zone_hints,
        )

    def ensure_route53_for_ingress(
        self, ingress, lb_ingress, hostnames: List[str], cluster_name: str,
        hint_arn: Optional[str] = None,
        zone_hints: Optional[dict] = None,
    ) -> Tuple[bool, float]:
        return self._ensure_route53(
            lb_ingress.hostname,
            hostnames,
            cluster_name,
            "ingress",
            ingress.metadata.namespace,
            ingress.metadata.name,
            hint_arn=hint_arn,
            zone_hints=zone_hints,
        )

    def _verified_hostname_hint(self, hint_arn: str, lb_hostname: str, cluster_name: str):
        from .global_accelerator import (
            GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY,
            GLOBAL_ACCELERATOR_MANAGED_TAG_KEY,
            GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY,
            tags_contains_all_values,
        )

        try:
            accelerator = self._get_accelerator(hint_arn)
            tags = self._list_tags_for_accelerator(hint_arn)
        except Exception:
            metrics.observe_hint("route53", "error")
            return None
        if tags_contains_all_values(
            tags,
            {
                GLOBAL_ACCELERATOR_MANAGED_TAG_KEY: "true",
                GLOBAL_ACCELERATOR_TARGET_HOSTNAME_KEY: lb_hostname,
                GLOBAL_ACCELERATOR_CLUSTER_TAG_KEY: cluster_name,
            },
        ):
            metrics.observe_hint("route53", "hit")
            return [accelerator]
        metrics.observe_hint("route53", "stale")
        return None

    def _ensure_route53(
        self,
        lb_hostname: str,
        hostnames: List[str],
        cluster_name: str,
        resource: str,
        ns: str,
        name: str,
        hint_arn: Optional[str] = None,
        zone_hints: Optional[dict] = None,
    ) -> Tuple[bool, float]:
        """Returns (created, retry_after_seconds).  0 or >1 matching
        accelerators ⇒ requeue after 60s (reference :62-78)."""
        accelerators = None
        if hint_arn:
            accelerators = self._verified_hostname_hint(
                hint_arn, lb_hostname, cluster_name
            )
        if accelerators is None:
            accelerators = self.list_global_accelerator_by_hostname(
                lb_hostname, cluster_name
            )
        if len(accelerators) > 1:
            logger.error("Too many Global Accelerators for %s", lb_hostname)
            return False, self.ga_missing_retry
        if not accelerators:
            logger.error("Could not find Global Accelerator for %s", lb_hostname)
            return False, self.ga_missing_retry
        accelerator = accelerators[0]
        # expose the matched accelerator so callers can maintain their hint
        # without a second scan (read via getattr right after the call)
        self._last_matched_accelerator_arn = accelerator.accelerator_arn

        owner_value = route53_owner_value(cluster_name, resource, ns, name)
        created = False
        import time as _time

        for hostname in hostnames:
            hinted = None
            if zone_hints is not None:
                entry = zone_hints.get(hostname)
                if entry is not None:
                    zone, stamp = entry
                    # TTL bounds divergence from the reference's
                    # per-reconcile walk: a newly-created more-specific
                    # hosted zone is picked up within ZONE_HINT_TTL
                    if _time.monotonic() - stamp < ZONE_HINT_TTL:
                        hinted = zone
                    else:
                        zone_hints.pop(hostname, None)
            hosted_zone = hinted or self.get_hosted_zone(hostname)
            logger.info("HostedZone is %s", hosted_zone.id)
            try:
                did_create = self._sync_one_hostname(
                    hosted_zone, hostname, owner_value, accelerator
                )
            except awserr.NoSuchHostedZone:
                if hinted is None:
                    raise
                # stale zone hint (zone deleted/recreated): drop it and
                # redo the reference's parent-domain walk once
                zone_hints.pop(hostname, None)
                hinted = None
                hosted_zone = self.get_hosted_zone(hostname)
                did_create = self._sync_one_hostname(
                    hosted_zone, hostname, owner_value, accelerator
                )
            created = created or did_create
            if zone_hints is not None and hinted is None:
                zone_hints[hostname] = (hosted_zone, _time.monotonic())

        logger.info("All records are synced for %s %s/%s", resource, ns, name)
        return created, 0.0

    def _sync_one_hostname(
        self, hosted_zone: t.HostedZone, hostname: str, owner_value: str, accelerator
    ) -> bool:
        """Create or drift-repair the TXT+A pair for one hostname in one
        zone; returns True if records were created."""
        record = self._find_owned_a_record_at(hosted_zone, hostname, owner_value)
        if record is None:
            logger.info(
                "Creating record for %s with %s",
                hostname,
                accelerator.accelerator_arn,
            )
            self._create_metadata_record_set(hosted_zone, hostname, owner_value)
            self._create_record_set(hosted_zone, hostname, accelerator)
            return True
        if not need_records_update(record, accelerator):
            logger.info("Do not need to update for %s, so skip it", record.name)
            return False
        self._update_record_set(hosted_zone, hostname, accelerator)
        logger.info("RecordSet %s is updated", record.name)
        return False

    def cleanup_record_set(
        self, cluster_name: str, resource: str, ns: str, name: str
    ):
        """Scan every zone; delete owned A-alias records then their TXT
        ownership records (reference :132-165)."""
        owner_value = route53_owner_value(cluster_name, resource, ns, name)
        for zone in self._list_all_hosted_zones():
            for record in self.find_owned_a_record_sets(zone, owner_value):
                self._delete_record(zone, record)
                logger.info("Record set %s: %s is deleted", record.name, record.type)
            for record in self._find_owned_metadata_record_sets(zone, owner_value):
                self._delete_record(zone, record)
                logger.info("Record set %s: %s is deleted", record.name, record.type)

    # -- record discovery ---------------------------------------------------
    def _find_owned_a_record_at(self, hosted_zone, hostname: str, owner_value: str):
        """Name-scoped variant of the reference's owned-record discovery
        (FindOwneredARecordSets + findARecord scan the whole zone,
        route53.go:216-238/360-367): list only the records AT ``hostname``
        via StartRecordName and decide from those.  The decision — create
        (no owned TXT), skip/update (owned TXT + alias A) — is identical;
        only the API cost drops from O(zone records) to O(1).  The cleanup
        paths still use the full-zone scan (they must find every owned
        name)."""
        target = hostname + "." if not hostname.endswith(".") else hostname
        target = target.replace("*", "\\052", 1)
        records_at_name: List[t.ResourceRecordSet] = []
        token = None
        while True:
            metrics.observe_aws_call("route53", "ListResourceRecordSets")
            page, token = self.route53.list_resource_record_sets(
                hosted_zone.id, max_items=10, start_token=token,
                start_record_name=hostname,
            )
            matched_this_page = False
            for rs in page:
                if rs.name == target:
                    records_at_name.append(rs)
                    matched_this_page = True
            # Route53 lists names in reversed-label DNS order, not plain
            # lexicographic order, so the only safe early exit is "this page
            # held no record at the target name" — StartRecordName positions
            # the scan at the name, so all matches are in a contiguous prefix.
            if not matched_this_page or token is None:
                break
        owned = any(
            record.value == owner_value
            for rs in records_at_name
            for record in rs.resource_records
        )
        if not owned:
            return None
        return next(
            (
                rs
                for rs in records_at_name
                if rs.type == t.RR_TYPE_A
                and rs.alias_target is not None
                and replace_wildcards(rs.name) == hostname + "."
            ),
            None,
        )

    def find_owned_a_record_sets(
        self, hosted_zone: t.HostedZone, owner_value: str
    ) -> List[t.ResourceRecordSet]:
        """Alias record sets whose name has a TXT record carrying our
        ownership value (reference ``FindOwneredARecordSets``, :216-238)."""
        record_sets = self._list_record_sets(hosted_zone.id)
        owned_names = [
            rs.name
            for rs in record_sets
            for record in rs.resource_records
            if record.value == owner_value
        ]
        return [
            rs
            for rs in record_sets
            if rs.name in owned_names and rs.alias_target is not None
        ]

    def _find_owned_metadata_record_sets(
        self, hosted_zone: t.HostedZone, owner_value: str
    ) -> List[t.ResourceRecordSet]:
        return [
            rs
            for rs in self._list_record_sets(hosted_zone.id)
            for record in rs.resource_records
            if record.value == owner_value
        ]

    # -- zone discovery ------------------------------------------------------
    def get_hosted_zone(self, original_hostname: str) -> t.HostedZone:
        """Walk up parent domains until a hosted zone matches
        (reference ``GetHostedZone``, :335-358)."""
        target = original_hostname
        while True:
            if not target:
                raise ValueError(
                    f"Could not find hosted zone for {original_hostname}"
                )
            logger.debug("Getting hosted zone for %s", target)
            metrics.observe_aws_call("route53", "ListHostedZonesByName")
            zones = self.route53.list_hosted_zones_by_name(
                dns_name=target + ".", max_items=1
            )
            for zone in zones:
                if zone.name == target + ".":
                    return zone
            target = parent_domain(target)

    def _list_all_hosted_zones(self) -> List[t.HostedZone]:
        zones: List[t.HostedZone] = []
        token = None
        while True:
            metrics.observe_aws_call("route53", "ListHostedZones")
            page, token = self.route53.list_hosted_zones(max_items=100, marker=token)
            zones.extend(page)
            if token is None:
                return zones

    def _list_record_sets(self, zone_id: str) -> List[t.ResourceRecordSet]:
        records: List[t.ResourceRecordSet] = []
        token = None
        while True:
            metrics.observe_aws_call("route53", "ListResourceRecordSets")
            page, token = self.route53.list_resource_record_sets(
                zone_id, max_items=300, start_token=token
            )
            records.extend(page)
            if token is None:
                return records

    # -- record mutations ----------------------------------------------------
    def _change(self, zone: t.HostedZone, action: str, record_set: t.ResourceRecordSet):
        metrics.observe_aws_call("route53", "ChangeResourceRecordSets")
        self.route53.change_resource_record_sets(
            zone.id, [t.Change(action=action, record_set=record_set)]
        )

    def _alias_record_set(self, hostname: str, accelerator) -> t.ResourceRecordSet:
        return t.ResourceRecordSet(
            name=hostname,
            type=t.RR_TYPE_A,
            alias_target=t.AliasTarget(
                dns_name=accelerator.dns_name,
                evaluate_target_health=True,
                # every Global Accelerator lives in this fixed alias zone
                hosted_zone_id=t.GLOBAL_ACCELERATOR_HOSTED_ZONE_ID,
            ),
        )

    def _create_record_set(self, zone: t.HostedZone, hostname: str, accelerator):
        try:
            self._change(
                zone, t.CHANGE_ACTION_CREATE, self._alias_record_set(hostname, accelerator)
            )
        except awserr.InvalidChangeBatch:
            # Partial-create recovery (deliberate fix, docs/PARITY.md): a
            # previous attempt may have committed the A record before a
            # transient failure.  This path is only reached after TXT
            # ownership of the name is established, so the A alias at the
            # name is ours — converge it with UPSERT instead of looping on
            # InvalidChangeBatch forever (the reference would loop:
            # createRecordSet always uses CREATE, route53.go:240-264).
            logger.info("A record for %s already exists; upserting", hostname)
            self._change(
                zone, t.CHANGE_ACTION_UPSERT, self._alias_record_set(hostname, accelerator)
            )

    def _update_record_set(self, zone: t.HostedZone, hostname: str, accelerator):
        self._change(zone, t.CHANGE_ACTION_UPSERT, self._alias_record_set(hostname, accelerator))

    def _create_metadata_record_set(
        self, zone: t.HostedZone, hostname: str, owner_value: str
    ):
        try:
            self._change(
                zone,
                t.CHANGE_ACTION_CREATE,
                t.ResourceRecordSet(
                    name=hostname,
                    type=t.RR_TYPE_TXT,
                    ttl=TXT_TTL,
                    resource_records=[t.ResourceRecord(value=owner_value)],
                ),
            )
        except awserr.InvalidChangeBatch:
            # Partial-create recovery: idempotent success ONLY if the
            # existing TXT carries our ownership value; anything else means
            # the name belongs to someone — re-raise (no stealing).
            target = hostname if hostname.endswith(".") else hostname + "."
            target = target.replace("*", "\\052", 1)
            page, _ = self.route53.list_resource_record_sets(
                zone.id, max_items=10, start_record_name=hostname
            )
            for rs in page:
                if (
                    rs.name == target
                    and rs.type == t.RR_TYPE_TXT
                    and any(r.value == owner_value for r in rs.resource_records)
                ):
                    logger.info("TXT ownership for %s already committed", hostname)
                    return
            raise

    def _delete_record(self, zone: t.HostedZone, record: t.ResourceRecordSet):
        self._change(zone, t.CHANGE_ACTION_DELETE, record)
