"""Hostname → (name, region) parsing parity
(reference pkg/cloudprovider/aws/load_balancer_test.go)."""

import pytest

from agac.cloudprovider import detect_cloud_provider
from agac.cloudprovider.aws import get_lb_name_from_hostname, get_region_from_arn


class TestALBHostnames:
    def test_public_alb(self):
        name, region = get_lb_name_from_hostname(
            "myapp-ingress-1234567890.ap-northeast-1.elb.amazonaws.com"
        )
        assert name == "myapp-ingress"
        assert region == "ap-northeast-1"

    def test_internal_alb(self):
        name, region = get_lb_name_from_hostname(
            "internal-myapp-ingress-1234567890.us-east-1.elb.amazonaws.com"
        )
        assert name == "myapp-ingress"
        assert region == "us-east-1"

    def test_alb_single_word_name(self):
        name, region = get_lb_name_from_hostname(
            "web-abc123.eu-west-1.elb.amazonaws.com"
        )
        assert name == "web"
        assert region == "eu-west-1"


class TestNLBHostnames:
    def test_nlb(self):
        name, region = get_lb_name_from_hostname(
            "myservice-nlb-0123456789abcdef.elb.ap-northeast-1.amazonaws.com"
        )
        assert name == "myservice-nlb"
        assert region == "ap-northeast-1"

    def test_nlb_simple(self):
        name, region = get_lb_name_from_hostname(
            "api-aabbccdd00112233.elb.us-west-2.amazonaws.com"
        )
        assert name == "api"
        assert region == "us-west-2"


class TestNonELB:
    @pytest.mark.parametrize(
        "hostname",
        [
            "example.com",
            "foo.cloudfront.net",
            "s3.amazonaws.com",
            "myapp.us-east-1.rds.amazonaws.com",
        ],
    )
    def test_rejected(self, hostname):
        with pytest.raises(ValueError):
            get_lb_name_from_hostname(hostname)

    def test_unparseable_subdomain(self):
        # no '-hash' suffix → parse failure
        with pytest.raises(ValueError):
            get_lb_name_from_hostname("justone.us-east-1.elb.amazonaws.com")


def test_get_region_from_arn():
    assert (
        get_region_from_arn(
            "arn:aws:elasticloadbalancing:ap-northeast-1:123456789012:loadbalancer/net/x/abc"
        )
        == "ap-northeast-1"
    )
    assert get_region_from_arn("arn:aws:globalaccelerator::123:accelerator/a") == ""


class TestDetectCloudProvider:
    def test_aws(self):
        assert (
            detect_cloud_provider("foo-123.elb.us-east-1.amazonaws.com") == "aws"
        )

    def test_unknown(self):
        with pytest.raises(ValueError):
            detect_cloud_provider("foo.example.org")

    def test_short_hostname(self):
        with pytest.raises(ValueError):
            detect_cloud_provider("localhost")


class TestReferenceHostnameTable:
    """The reference's exact 4-case table (load_balancer_test.go:15-41),
    with its real-world hostname shapes."""

    def test_public_nlb_hex_name(self):
        name, region = get_lb_name_from_hostname(
            "aa5849cde256f49faa7487bb433155b7-3f43353a6cb6f633.elb.ap-northeast-1.amazonaws.com"
        )
        assert name == "aa5849cde256f49faa7487bb433155b7"
        assert region == "ap-northeast-1"

    def test_internal_nlb(self):
        name, region = get_lb_name_from_hostname(
            "test-b6cdc5fbd1d6fa43.elb.ap-northeast-1.amazonaws.com"
        )
        assert name == "test"
        assert region == "ap-northeast-1"

    def test_public_alb_k8s_style(self):
        name, region = get_lb_name_from_hostname(
            "k8s-default-h3poteto-f1f41628db-201899272.ap-northeast-1.elb.amazonaws.com"
        )
        assert name == "k8s-default-h3poteto-f1f41628db"
        assert region == "ap-northeast-1"

    def test_internal_alb_k8s_style(self):
        name, region = get_lb_name_from_hostname(
            "internal-k8s-default-h3poteto-35ca57562f-777774719.ap-northeast-1.elb.amazonaws.com"
        )
        assert name == "k8s-default-h3poteto-35ca57562f"
        assert region == "ap-northeast-1"
