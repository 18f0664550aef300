"""boto3-backed service clients (production path).

Adapts boto3's elbv2 / globalaccelerator / route53 clients to the operation
protocol the resource managers consume (the same surface the in-memory fake
implements), translating botocore ClientError codes into this package's
typed errors.  The reference builds the equivalent SDK clients in
``pkg/cloudprovider/aws/aws.go:18-38`` — ELBv2 regional, Global Accelerator
and Route53 pinned to us-west-2 (GA is a global service homed there).

boto3 is an optional dependency (``pip install agac[aws]``); importing this
module without it raises ImportError.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import boto3
from botocore.exceptions import ClientError

from . import errors as awserr
from . import types as t
from .client import AWS

_ERROR_CLASSES = {
    cls.code: cls
    for cls in (
        awserr.AcceleratorNotFoundException,
        awserr.ListenerNotFoundException,
        awserr.EndpointGroupNotFoundException,
        awserr.LoadBalancerNotFoundException,
        awserr.AcceleratorNotDisabledException,
        awserr.NoSuchHostedZone,
        awserr.InvalidChangeBatch,
    )
}


def _translate(err: ClientError) -> awserr.AWSAPIError:
    code = err.response.get("Error", {}).get("Code", "InternalError")
    message = err.response.get("Error", {}).get("Message", str(err))
    cls = _ERROR_CLASSES.get(code)
    if cls is not None:
        return cls(message)
    return awserr.AWSAPIError(message, code)


def _call(fn, **kwargs):
    try:
        return fn(**kwargs)
    except ClientError as e:
        raise _translate(e) from e


def _tags_out(tags) -> List[t.Tag]:
    return [t.Tag(key=x["Key"], value=x["Value"]) for x in tags or []]


def _tags_in(tags: List[t.Tag]):
    return [{"Key": x.key, "Value": x.value} for x in tags]


class Boto3ELBv2:
    def __init__(self, client):
        self.client = client

    def describe_load_balancers(
        self,
        names: Optional[List[str]] = None,
        marker: Optional[str] = None,
        page_size: Optional[int] = None,
    ) -> Tuple[List[t.LoadBalancer], Optional[str]]:
        kwargs = {}
        if names:
            kwargs["Names"] = names
        if marker:
            kwargs["Marker"] = marker
        if page_size:
            kwargs["PageSize"] = page_size
        res = _call(self.client.describe_load_balancers, **kwargs)
        lbs = [
            t.LoadBalancer(
                load_balancer_arn=lb["LoadBalancerArn"],
                load_balancer_name=lb["LoadBalancerName"],
                dns_name=lb.get("DNSName", ""),
                state_code=(lb.get("State") or {}).get("Code", ""),
                type=lb.get("Type", ""),
                scheme=lb.get("Scheme", ""),
            )
            for lb in res.get("LoadBalancers", [])
        ]
        return lbs, res.get("NextMarker")


class Boto3GlobalAccelerator:
    def __init__(self, client):
        self.client = client

    # -- accelerators ------------------------------------------------------
    def _acc_out(self, a) -> t.Accelerator:
        return t.Accelerator(
            accelerator_arn=a["AcceleratorArn"],
            name=a.get("Name", ""),
            dns_name=a.get("DnsName", ""),
            enabled=a.get("Enabled", False),
            status=a.get("Status", ""),
            ip_address_type=a.get("IpAddressType", ""),
        )

    def create_accelerator(self, name, ip_address_type, enabled, tags):
        res = _call(
            self.client.create_accelerator,
            Name=name,
            IpAddressType=ip_address_type,
            Enabled=enabled,
            Tags=_tags_in(tags or []),
        )
        return self._acc_out(res["Accelerator"])

    def describe_accelerator(self, arn):
        res = _call(self.client.describe_accelerator, AcceleratorArn=arn)
        return self._acc_out(res["Accelerator"])

    def list_accelerators(self, max_results=None, next_token=None):
        kwargs = {}
        if max_results:
            kwargs["MaxResults"] = max_results
        if next_token:
            kwargs["NextToken"] = next_token
        res = _call(self.client.list_accelerators, **kwargs)
        return [self._acc_out(a) for a in res.get("Accelerators", [])], res.get("NextToken")

    def update_accelerator(self, arn, name=None, enabled=None, ip_address_type=None):
        kwargs = {"AcceleratorArn": arn}
        if name is not None:
            kwargs["Name"] = name
        if enabled is not None:
            kwargs["Enabled"] = enabled
        if ip_address_type is not None:
            kwargs["IpAddressType"] = ip_address_type
        res = _call(self.client.update_accelerator, **kwargs)
        return self._acc_out(res["Accelerator"])

    def delete_accelerator(self, arn):
        _call(self.client.delete_accelerator, AcceleratorArn=arn)

    def list_tags_for_resource(self, arn):
        res = _call(self.client.list_tags_for_resource, ResourceArn=arn)
        return _tags_out(res.get("Tags"))

    def tag_resource(self, arn, tags):
        _call(self.client.tag_resource, ResourceArn=arn, Tags=_tags_in(tags))

    # -- listeners ---------------------------------------------------------
    def _listener_out(self, l) -> t.Listener:
        return t.Listener(
            listener_arn=l["ListenerArn"],
            port_ranges=[
                t.PortRange(from_port=p["FromPort"], to_port=p["ToPort"])
                for p in l.get("PortRanges", [])
            ],
            protocol=l.get("Protocol", ""),
            client_affinity=l.get("ClientAffinity", ""),
        )

    @staticmethod
    def _ranges_in(port_ranges):
        return [{"FromPort": p.from_port, "ToPort": p.to_port} for p in port_ranges]

    def create_listener(self, accelerator_arn, port_ranges, protocol, client_affinity="NONE"):
        res = _call(
            self.client.create_listener,
            AcceleratorArn=accelerator_arn,
            PortRanges=self._ranges_in(port_ranges),
            Protocol=protocol,
            ClientAffinity=client_affinity,
        )
        return self._listener_out(res["Listener"])

    def list_listeners(self, accelerator_arn, max_results=None, next_token=None):
        kwargs = {"AcceleratorArn": accelerator_arn}
        if max_results:
            kwargs["MaxResults"] = max_results
        if next_token:
            kwargs["NextToken"] = next_token
        res = _call(self.client.list_listeners, **kwargs)
        return [self._listener_out(l) for l in res.get("Listeners", [])], res.get("NextToken")

    def update_listener(self, listener_arn, port_ranges=None, protocol=None, client_affinity=None):
        kwargs = {"ListenerArn": listener_arn}
        if port_ranges is not None:
            kwargs["PortRanges"] = self._ranges_in(port_ranges)
        if protocol is not None:
            kwargs["Protocol"] = protocol
        if client_affinity is not None:
            kwargs["ClientAffinity"] = client_affinity
        res = _call(self.client.update_listener, **kwargs)
        return self._listener_out(res["Listener"])

    def delete_listener(self, listener_arn):
        _call(self.client.delete_listener, ListenerArn=listener_arn)

    # -- endpoint groups ---------------------------------------------------
    def _eg_out(self, g) -> t.EndpointGroup:
        return t.EndpointGroup(
            endpoint_group_arn=g["EndpointGroupArn"],
            endpoint_group_region=g.get("EndpointGroupRegion", ""),
            endpoint_descriptions=[
                t.EndpointDescription(
                    endpoint_id=d.get("EndpointId", ""),
                    weight=d.get("Weight"),
                    client_ip_preservation_enabled=d.get("ClientIPPreservationEnabled"),
                    health_state=d.get("HealthState", ""),
                )
                for d in g.get("EndpointDescriptions", [])
            ],
        )

    @staticmethod
    def _configs_in(configs):
        out = []
        for c in configs or []:
            entry = {"EndpointId": c.endpoint_id}
            if c.weight is not None:
                entry["Weight"] = c.weight
            if c.client_ip_preservation_enabled is not None:
                entry["ClientIPPreservationEnabled"] = c.client_ip_preservation_enabled
            out.append(entry)
        return out

    def create_endpoint_group(self, listener_arn, endpoint_group_region, endpoint_configurations=None):
        res = _call(
            self.client.create_endpoint_group,
            ListenerArn=listener_arn,
            EndpointGroupRegion=endpoint_group_region,
            EndpointConfigurations=self._configs_in(endpoint_configurations),
        )
        return self._eg_out(res["EndpointGroup"])

    def list_endpoint_groups(self, listener_arn, max_results=None, next_token=None):
        kwargs = {"ListenerArn": listener_arn}
        if max_results:
            kwargs["MaxResults"] = max_results
        if next_token:
            kwargs["NextToken"] = next_token
        res = _call(self.client.list_endpoint_groups, **kwargs)
        return [self._eg_out(g) for g in res.get("EndpointGroups", [])], res.get("NextToken")

    def describe_endpoint_group(self, endpoint_group_arn):
        res = _call(self.client.describe_endpoint_group, EndpointGroupArn=endpoint_group_arn)
        return self._eg_out(res["EndpointGroup"])

    def update_endpoint_group(self, endpoint_group_arn, endpoint_configurations=None):
        kwargs = {"EndpointGroupArn": endpoint_group_arn}
        if endpoint_configurations is not None:
            kwargs["EndpointConfigurations"] = self._configs_in(endpoint_configurations)
        res = _call(self.client.update_endpoint_group, **kwargs)
        return self._eg_out(res["EndpointGroup"])

    def add_endpoints(self, endpoint_group_arn, endpoint_configurations):
        res = _call(
            self.client.add_endpoints,
            EndpointGroupArn=endpoint_group_arn,
            EndpointConfigurations=self._configs_in(endpoint_configurations),
        )
        return [
            t.EndpointDescription(
                endpoint_id=d.get("EndpointId", ""),
                weight=d.get("Weight"),
                client_ip_preservation_enabled=d.get("ClientIPPreservationEnabled"),
            )
            for d in res.get("EndpointDescriptions", [])
        ]

    def remove_endpoints(self, endpoint_group_arn, endpoint_ids):
        _call(
            self.client.remove_endpoints,
            EndpointGroupArn=endpoint_group_arn,
            EndpointIdentifiers=[{"EndpointId": e} for e in endpoint_ids],
        )

    def delete_endpoint_group(self, endpoint_group_arn):
        _call(self.client.delete_endpoint_group, EndpointGroupArn=endpoint_group_arn)


class Boto3Route53:
    def __init__(self, client):
        self.client = client

    def _zone_out(self, z) -> t.HostedZone:
        return t.HostedZone(id=z["Id"], name=z.get("Name", ""))

    def _record_out(self, r) -> t.ResourceRecordSet:
        alias = r.get("AliasTarget")
        return t.ResourceRecordSet(
            name=r.get("Name", ""),
            type=r.get("Type", ""),
            ttl=r.get("TTL"),
            resource_records=[
                t.ResourceRecord(value=v.get("Value", ""))
                for v in r.get("ResourceRecords", [])
            ],
            alias_target=t.AliasTarget(
                dns_name=alias.get("DNSName", ""),
                evaluate_target_health=alias.get("EvaluateTargetHealth", False),
                hosted_zone_id=alias.get("HostedZoneId", ""),
            )
            if alias
            else None,
        )

    def _record_in(self, r: t.ResourceRecordSet) -> dict:
        out = {"Name": r.name, "Type": r.type}
        if r.ttl is not None:
            out["TTL"] = r.ttl
        if r.resource_records:
            out["ResourceRecords"] = [{"Value": x.value} for x in r.resource_records]
        if r.alias_target is not None:
            out["AliasTarget"] = {
                "DNSName": r.alias_target.dns_name,
                "EvaluateTargetHealth": r.alias_target.evaluate_target_health,
                "HostedZoneId": r.alias_target.hosted_zone_id,
            }
        return out

    def list_hosted_zones(self, max_items=None, marker=None):
        kwargs = {}
        if max_items:
            kwargs["MaxItems"] = str(max_items)
        if marker:
            kwargs["Marker"] = marker
        res = _call(self.client.list_hosted_zones, **kwargs)
        zones = [self._zone_out(z) for z in res.get("HostedZones", [])]
        return zones, res.get("NextMarker") if res.get("IsTruncated") else None

    def list_hosted_zones_by_name(self, dns_name, max_items=None):
        kwargs = {"DNSName": dns_name}
        if max_items:
            kwargs["MaxItems"] = str(max_items)
        res = _call(self.client.list_hosted_zones_by_name, **kwargs)
        return [self._zone_out(z) for z in res.get("HostedZones", [])]

    def list_resource_record_sets(self, zone_id, max_items=None, start_token=None,
                                  start_record_name=None):
        kwargs = {"HostedZoneId": zone_id}
        if max_items:
            kwargs["MaxItems"] = str(max_items)
        if start_token:
            name, rtype = start_token.split("|", 1)
            kwargs["StartRecordName"] = name
            kwargs["StartRecordType"] = rtype
        elif start_record_name:
            kwargs["StartRecordName"] = start_record_name
        res = _call(self.client.list_resource_record_sets, **kwargs)
        records = [self._record_out(r) for r in res.get("ResourceRecordSets", [])]
        token = None
        if res.get("IsTruncated"):
            token = f'{res.get("NextRecordName", "")}|{res.get("NextRecordType", "")}'
        return records, token

    def change_resource_record_sets(self, zone_id, changes):
        _call(
            self.client.change_resource_record_sets,
            HostedZoneId=zone_id,
            ChangeBatch={
                "Changes": [
                    {"Action": c.action, "ResourceRecordSet": self._record_in(c.record_set)}
                    for c in changes
                ]
            },
        )


def new_boto3_factory():
    """CloudFactory over boto3 (reference NewAWS(region), aws.go:18-38):
    ELBv2 regional; Global Accelerator + Route53 pinned to us-west-2."""
    session = boto3.session.Session()
    ga = Boto3GlobalAccelerator(session.client("globalaccelerator", region_name="us-west-2"))
    route53 = Boto3Route53(session.client("route53", region_name="us-west-2"))

    def factory(region: str) -> AWS:
        lb = Boto3ELBv2(session.client("elbv2", region_name=region))
        return AWS(lb=lb, ga=ga, route53=route53, region=region)

    return factory
