"""boto3 adapter tests against stubbed boto3/botocore modules: request
kwargs translation, response shaping, pagination tokens, error-code
translation, and the region pinning of new_boto3_factory — the production
seam verified without AWS or boto3 installed."""

import sys
import types

import pytest


class StubClientError(Exception):
    def __init__(self, code, message="boom"):
        self.response = {"Error": {"Code": code, "Message": message}}
        super().__init__(message)


class RecordingClient:
    """Returns canned responses per method name; records every call."""

    def __init__(self, responses=None):
        self.responses = responses or {}
        self.calls = []

    def __getattr__(self, method):
        def call(**kwargs):
            self.calls.append((method, kwargs))
            result = self.responses.get(method)
            if isinstance(result, Exception):
                raise result
            if callable(result):
                return result(**kwargs)
            return result or {}

        return call


@pytest.fixture()
def adapter(monkeypatch):
    """Import agac.cloudprovider.aws.boto3_adapter against stub modules."""
    boto3_stub = types.ModuleType("boto3")
    session_mod = types.ModuleType("boto3.session")

    created_clients = {}

    class Session:
        def client(self, service, region_name=None):
            client = RecordingClient()
            created_clients.setdefault(service, []).append((region_name, client))
            return client

    session_mod.Session = Session
    boto3_stub.session = session_mod

    botocore_stub = types.ModuleType("botocore")
    exceptions_mod = types.ModuleType("botocore.exceptions")
    exceptions_mod.ClientError = StubClientError
    botocore_stub.exceptions = exceptions_mod

    monkeypatch.setitem(sys.modules, "boto3", boto3_stub)
    monkeypatch.setitem(sys.modules, "boto3.session", session_mod)
    monkeypatch.setitem(sys.modules, "botocore", botocore_stub)
    monkeypatch.setitem(sys.modules, "botocore.exceptions", exceptions_mod)

    sys.modules.pop("agac.cloudprovider.aws.boto3_adapter", None)
    import agac.cloudprovider.aws.boto3_adapter as mod

    yield mod, created_clients
    sys.modules.pop("agac.cloudprovider.aws.boto3_adapter", None)


class TestELBv2Adapter:
    def test_describe_mapping(self, adapter):
        mod, _ = adapter
        client = RecordingClient(
            {
                "describe_load_balancers": {
                    "LoadBalancers": [
                        {
                            "LoadBalancerArn": "arn:lb",
                            "LoadBalancerName": "web",
                            "DNSName": "web-1.elb.us-east-1.amazonaws.com",
                            "State": {"Code": "active"},
                            "Type": "network",
                            "Scheme": "internet-facing",
                        }
                    ],
                    "NextMarker": "m2",
                }
            }
        )
        lbs, marker = mod.Boto3ELBv2(client).describe_load_balancers(
            names=["web"], page_size=5
        )
        assert client.calls == [
            ("describe_load_balancers", {"Names": ["web"], "PageSize": 5})
        ]
        assert lbs[0].load_balancer_arn == "arn:lb"
        assert lbs[0].state_code == "active"
        assert marker == "m2"

    def test_error_translation(self, adapter):
        mod, _ = adapter
        from agac.cloudprovider.aws.errors import LoadBalancerNotFoundException

        client = RecordingClient(
            {"describe_load_balancers": StubClientError("LoadBalancerNotFound")}
        )
        with pytest.raises(LoadBalancerNotFoundException):
            mod.Boto3ELBv2(client).describe_load_balancers(names=["ghost"])


class TestGlobalAcceleratorAdapter:
    def test_create_accelerator_tags(self, adapter):
        mod, _ = adapter
        from agac.cloudprovider.aws import types as t

        client = RecordingClient(
            {
                "create_accelerator": {
                    "Accelerator": {
                        "AcceleratorArn": "arn:acc",
                        "Name": "n",
                        "DnsName": "d",
                        "Enabled": True,
                        "Status": "IN_PROGRESS",
                        "IpAddressType": "DUAL_STACK",
                    }
                }
            }
        )
        acc = mod.Boto3GlobalAccelerator(client).create_accelerator(
            "n", "DUAL_STACK", True, [t.Tag("k", "v")]
        )
        method, kwargs = client.calls[0]
        assert kwargs["Tags"] == [{"Key": "k", "Value": "v"}]
        assert acc.accelerator_arn == "arn:acc"
        assert acc.status == "IN_PROGRESS"

    def test_endpoint_group_roundtrip(self, adapter):
        mod, _ = adapter
        from agac.cloudprovider.aws import types as t

        client = RecordingClient(
            {
                "update_endpoint_group": {
                    "EndpointGroup": {
                        "EndpointGroupArn": "arn:eg",
                        "EndpointGroupRegion": "us-east-1",
                        "EndpointDescriptions": [
                            {"EndpointId": "arn:lb", "Weight": 7,
                             "ClientIPPreservationEnabled": True,
                             "HealthState": "HEALTHY"}
                        ],
                    }
                }
            }
        )
        eg = mod.Boto3GlobalAccelerator(client).update_endpoint_group(
            "arn:eg",
            endpoint_configurations=[
                t.EndpointConfiguration(endpoint_id="arn:lb", weight=7,
                                        client_ip_preservation_enabled=True)
            ],
        )
        method, kwargs = client.calls[0]
        assert kwargs["EndpointConfigurations"] == [
            {"EndpointId": "arn:lb", "Weight": 7, "ClientIPPreservationEnabled": True}
        ]
        assert eg.endpoint_descriptions[0].weight == 7

    def test_typed_not_found(self, adapter):
        mod, _ = adapter
        from agac.cloudprovider.aws.errors import EndpointGroupNotFoundException

        client = RecordingClient(
            {"describe_endpoint_group": StubClientError("EndpointGroupNotFoundException")}
        )
        with pytest.raises(EndpointGroupNotFoundException):
            mod.Boto3GlobalAccelerator(client).describe_endpoint_group("arn:x")

    def test_pagination_token_passthrough(self, adapter):
        mod, _ = adapter
        client = RecordingClient(
            {"list_accelerators": {"Accelerators": [], "NextToken": "t2"}}
        )
        items, token = mod.Boto3GlobalAccelerator(client).list_accelerators(
            max_results=100, next_token="t1"
        )
        assert client.calls[0][1] == {"MaxResults": 100, "NextToken": "t1"}
        assert token == "t2"


class TestRoute53Adapter:
    def test_change_batch_shape(self, adapter):
        mod, _ = adapter
        from agac.cloudprovider.aws import types as t

        client = RecordingClient({"change_resource_record_sets": {}})
        mod.Boto3Route53(client).change_resource_record_sets(
            "Z1",
            [
                t.Change(
                    action="CREATE",
                    record_set=t.ResourceRecordSet(
                        name="a.example.com", type="A",
                        alias_target=t.AliasTarget(
                            dns_name="d.", evaluate_target_health=True,
                            hosted_zone_id="Z2BJ6XQ5FK7U4H",
                        ),
                    ),
                )
            ],
        )
        method, kwargs = client.calls[0]
        assert kwargs["HostedZoneId"] == "Z1"
        change = kwargs["ChangeBatch"]["Changes"][0]
        assert change["Action"] == "CREATE"
        assert change["ResourceRecordSet"]["AliasTarget"]["HostedZoneId"] == "Z2BJ6XQ5FK7U4H"

    def test_record_list_truncation(self, adapter):
        mod, _ = adapter
        client = RecordingClient(
            {
                "list_resource_record_sets": {
                    "ResourceRecordSets": [
                        {"Name": "a.example.com.", "Type": "TXT", "TTL": 300,
                         "ResourceRecords": [{"Value": '"owner"'}]}
                    ],
                    "IsTruncated": True,
                    "NextRecordName": "b.example.com.",
                    "NextRecordType": "A",
                }
            }
        )
        records, token = mod.Boto3Route53(client).list_resource_record_sets("Z1", max_items=1)
        assert records[0].resource_records[0].value == '"owner"'
        assert token == "b.example.com.|A"
        # the token feeds back as StartRecordName/Type
        client.calls.clear()
        mod.Boto3Route53(client).list_resource_record_sets("Z1", start_token=token)
        assert client.calls[0][1]["StartRecordName"] == "b.example.com."
        assert client.calls[0][1]["StartRecordType"] == "A"

    def test_start_record_name_passthrough(self, adapter):
        mod, _ = adapter
        client = RecordingClient({"list_resource_record_sets": {"ResourceRecordSets": []}})
        mod.Boto3Route53(client).list_resource_record_sets(
            "Z1", start_record_name="www.example.com"
        )
        assert client.calls[0][1]["StartRecordName"] == "www.example.com"


class TestFactoryRegionPinning:
    def test_ga_and_route53_pinned_elbv2_regional(self, adapter):
        mod, created = adapter
        factory = mod.new_boto3_factory()
        cloud = factory("eu-central-1")
        # reference aws.go:18-38: GA + Route53 in us-west-2, ELB regional
        assert created["globalaccelerator"][0][0] == "us-west-2"
        assert created["route53"][0][0] == "us-west-2"
        assert created["elbv2"][0][0] == "eu-central-1"
        assert cloud.region == "eu-central-1"
        # a second region creates a fresh elbv2 client but reuses ga/route53
        factory("ap-northeast-1")
        assert len(created["elbv2"]) == 2
        assert len(created["globalaccelerator"]) == 1
