"""Route53 pure-helper parity tests
(reference pkg/cloudprovider/aws/route53_test.go, 183 LoC)."""

from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.route53 import (
    find_a_record,
    need_records_update,
    parent_domain,
    replace_wildcards,
    route53_owner_value,
)


def alias_record(name, target="abc.awsglobalaccelerator.com."):
    return t.ResourceRecordSet(
        name=name,
        type=t.RR_TYPE_A,
        alias_target=t.AliasTarget(dns_name=target, hosted_zone_id="Z2BJ6XQ5FK7U4H"),
    )


class TestOwnerValue:
    def test_format(self):
        assert route53_owner_value("c1", "service", "ns", "web") == (
            '"heritage=aws-global-accelerator-controller,cluster=c1,service/ns/web"'
        )


class TestParentDomain:
    def test_walks_up(self):
        assert parent_domain("a.b.example.com") == "b.example.com"
        assert parent_domain("example.com") == "com"
        assert parent_domain("com") == ""

    def test_trailing_dot_edge(self):
        # "." splits into two empty labels → parent is "" (reference edge case)
        assert parent_domain(".") == ""


class TestReplaceWildcards:
    def test_escaped(self):
        assert replace_wildcards("\\052.example.com.") == "*.example.com."

    def test_plain(self):
        assert replace_wildcards("www.example.com.") == "www.example.com."

    def test_only_first(self):
        assert replace_wildcards("\\052.\\052.com.") == "*.\\052.com."


class TestFindARecord:
    def test_exact_match(self):
        records = [alias_record("www.example.com.")]
        assert find_a_record(records, "www.example.com") is records[0]

    def test_wildcard_match(self):
        records = [alias_record("\\052.example.com.")]
        assert find_a_record(records, "*.example.com") is records[0]

    def test_no_match(self):
        records = [alias_record("www.example.com.")]
        assert find_a_record(records, "api.example.com") is None

    def test_ignores_txt(self):
        records = [
            t.ResourceRecordSet(name="www.example.com.", type=t.RR_TYPE_TXT)
        ]
        assert find_a_record(records, "www.example.com") is None


class TestNeedRecordsUpdate:
    def acc(self, dns="abc.awsglobalaccelerator.com"):
        return t.Accelerator(dns_name=dns)

    def test_up_to_date(self):
        record = alias_record("www.example.com.", target="abc.awsglobalaccelerator.com.")
        assert not need_records_update(record, self.acc())

    def test_drifted(self):
        record = alias_record("www.example.com.", target="old.awsglobalaccelerator.com.")
        assert need_records_update(record, self.acc())

    def test_missing_alias(self):
        record = t.ResourceRecordSet(name="www.example.com.", type=t.RR_TYPE_A)
        assert need_records_update(record, self.acc())
