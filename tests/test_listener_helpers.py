"""Listener derivation and drift predicates
(reference pkg/cloudprovider/aws/global_accelerator_test.go, 489 LoC of
table-driven cases)."""

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.cloudprovider.aws import types as t
from agac.cloudprovider.aws.global_accelerator import (
    accelerator_name,
    accelerator_owner_tag_value,
    accelerator_tags,
    endpoint_contains_lb,
    listener_for_ingress,
    listener_for_service,
    listener_port_changed_from_service,
    listener_protocol_changed_from_ingress,
    listener_protocol_changed_from_service,
    resolve_ip_address_type,
    tags_contains_all_values,
)


def svc_with_ports(*ports):
    return corev1.Service(
        metadata=ObjectMeta(name="web", namespace="default"),
        spec=corev1.ServiceSpec(
            type="LoadBalancer",
            ports=[corev1.ServicePort(port=p, protocol=proto) for p, proto in ports],
        ),
    )


def listener(ports, protocol=t.PROTOCOL_TCP):
    return t.Listener(
        listener_arn="arn:listener",
        port_ranges=[t.PortRange(from_port=p, to_port=p) for p in ports],
        protocol=protocol,
    )


class TestListenerForService:
    def test_tcp_ports(self):
        ports, protocol = listener_for_service(svc_with_ports((80, "TCP"), (443, "TCP")))
        assert ports == [80, 443]
        assert protocol == t.PROTOCOL_TCP

    def test_udp_wins_when_last(self):
        ports, protocol = listener_for_service(svc_with_ports((53, "TCP"), (53, "UDP")))
        assert protocol == t.PROTOCOL_UDP

    def test_last_protocol_wins(self):
        # reference :503-515 keeps the protocol of the last typed port
        _, protocol = listener_for_service(svc_with_ports((53, "UDP"), (80, "TCP")))
        assert protocol == t.PROTOCOL_TCP


class TestListenerProtocolChange:
    def test_unchanged(self):
        svc = svc_with_ports((80, "TCP"))
        assert not listener_protocol_changed_from_service(listener([80]), svc)

    def test_changed_to_udp(self):
        svc = svc_with_ports((53, "UDP"))
        assert listener_protocol_changed_from_service(listener([53]), svc)

    def test_ingress_requires_tcp(self):
        ingress = mk_ingress()
        assert not listener_protocol_changed_from_ingress(listener([80]), ingress)
        assert listener_protocol_changed_from_ingress(
            listener([80], protocol=t.PROTOCOL_UDP), ingress
        )


class TestListenerPortChanged:
    def test_same_ports(self):
        svc = svc_with_ports((80, "TCP"), (443, "TCP"))
        assert not listener_port_changed_from_service(listener([80, 443]), svc)

    def test_extra_listener_port(self):
        svc = svc_with_ports((80, "TCP"))
        assert listener_port_changed_from_service(listener([80, 443]), svc)

    def test_extra_service_port(self):
        svc = svc_with_ports((80, "TCP"), (8080, "TCP"))
        assert listener_port_changed_from_service(listener([80]), svc)

    def test_disjoint(self):
        svc = svc_with_ports((9000, "TCP"))
        assert listener_port_changed_from_service(listener([80]), svc)


def mk_ingress(annotations=None, default_port=None, rule_ports=()):
    spec = corev1.IngressSpec(ingress_class_name="alb")
    if default_port is not None:
        spec.default_backend = corev1.IngressBackend(
            service=corev1.IngressServiceBackend(
                name="default", port=corev1.ServiceBackendPort(number=default_port)
            )
        )
    if rule_ports:
        spec.rules = [
            corev1.IngressRule(
                host="x",
                http=corev1.HTTPIngressRuleValue(
                    paths=[
                        corev1.HTTPIngressPath(
                            backend=corev1.IngressBackend(
                                service=corev1.IngressServiceBackend(
                                    name="svc",
                                    port=corev1.ServiceBackendPort(number=p),
                                )
                            )
                        )
                        for p in rule_ports
                    ]
                ),
            )
        ]
    return corev1.Ingress(
        metadata=ObjectMeta(
            name="ing", namespace="default", annotations=annotations or {}
        ),
        spec=spec,
    )


class TestListenerForIngress:
    def test_listen_ports_annotation(self):
        ingress = mk_ingress(
            annotations={
                "alb.ingress.kubernetes.io/listen-ports": '[{"HTTP": 80}, {"HTTPS": 443}]'
            }
        )
        ports, protocol = listener_for_ingress(ingress)
        assert ports == [80, 443]
        assert protocol == t.PROTOCOL_TCP

    def test_listen_ports_both_in_one_entry(self):
        ingress = mk_ingress(
            annotations={
                "alb.ingress.kubernetes.io/listen-ports": '[{"HTTP": 80, "HTTPS": 443}]'
            }
        )
        ports, _ = listener_for_ingress(ingress)
        assert ports == [80, 443]

    def test_invalid_annotation_returns_empty(self):
        ingress = mk_ingress(
            annotations={"alb.ingress.kubernetes.io/listen-ports": "not-json"}
        )
        ports, _ = listener_for_ingress(ingress)
        assert ports == []

    def test_backend_ports_without_annotation(self):
        ingress = mk_ingress(default_port=8080, rule_ports=(80, 9090))
        ports, _ = listener_for_ingress(ingress)
        assert ports == [8080, 80, 9090]

    def test_no_ports(self):
        ports, _ = listener_for_ingress(mk_ingress())
        assert ports == []


class TestAcceleratorHelpers:
    def test_owner_tag_value(self):
        assert accelerator_owner_tag_value("service", "ns", "n") == "service/ns/n"

    def test_accelerator_name_default(self):
        svc = svc_with_ports((80, "TCP"))
        assert accelerator_name("service", svc) == "service-default-web"

    def test_accelerator_name_annotation(self):
        svc = svc_with_ports((80, "TCP"))
        svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-name"
        ] = "custom"
        assert accelerator_name("service", svc) == "custom"

    def test_accelerator_tags_parsing(self):
        svc = svc_with_ports((80, "TCP"))
        svc.metadata.annotations[
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-tags"
        ] = "env=prod,team=infra,malformed,also=ok"
        tags = accelerator_tags(svc)
        assert [(x.key, x.value) for x in tags] == [
            ("env", "prod"),
            ("team", "infra"),
            ("also", "ok"),
        ]

    def test_tags_contains_all_values(self):
        tags = [t.Tag("a", "1"), t.Tag("b", "2")]
        assert tags_contains_all_values(tags, {"a": "1"})
        assert not tags_contains_all_values(tags, {"a": "2"})
        assert not tags_contains_all_values(tags, {"c": "3"})

    def test_resolve_ip_address_type(self):
        assert resolve_ip_address_type("") == "DUAL_STACK"
        assert resolve_ip_address_type("ipv4") == "IPV4"
        assert resolve_ip_address_type("IPV4") == "IPV4"
        assert resolve_ip_address_type("dualstack") == "DUAL_STACK"
        assert resolve_ip_address_type("DUAL_STACK") == "DUAL_STACK"
        assert resolve_ip_address_type("bogus") == "DUAL_STACK"

    def test_endpoint_contains_lb(self):
        lb = t.LoadBalancer(load_balancer_arn="arn:lb")
        group = t.EndpointGroup(
            endpoint_descriptions=[t.EndpointDescription(endpoint_id="arn:lb")]
        )
        assert endpoint_contains_lb(group, lb)
        assert not endpoint_contains_lb(t.EndpointGroup(), lb)


class TestProtocolChangeReferenceTable:
    """The six cases of the reference's TestListenerProtocolChange table
    (global_accelerator_test.go:15-155), 1:1 — last-typed-port-wins decides."""

    def test_not_changed_single(self):
        assert not listener_protocol_changed_from_service(
            listener([80]), svc_with_ports((80, "TCP"))
        )

    def test_not_changed_multiple_same(self):
        assert not listener_protocol_changed_from_service(
            listener([80]), svc_with_ports((80, "TCP"), (443, "TCP"))
        )

    def test_not_changed_multiple_different_tcp_last(self):
        # UDP then TCP: last port's protocol (TCP) matches the listener
        assert not listener_protocol_changed_from_service(
            listener([80]), svc_with_ports((53, "UDP"), (80, "TCP"))
        )

    def test_changed_single(self):
        assert listener_protocol_changed_from_service(
            listener([53]), svc_with_ports((53, "UDP"))
        )

    def test_changed_multiple_udp(self):
        assert listener_protocol_changed_from_service(
            listener([53]), svc_with_ports((53, "UDP"), (54, "UDP"))
        )

    def test_changed_multiple_different_udp_last(self):
        # TCP then UDP: last is UDP, listener TCP → drift
        assert listener_protocol_changed_from_service(
            listener([80]), svc_with_ports((80, "TCP"), (53, "UDP"))
        )


class TestPortChangeReferenceTable:
    """Remaining cases of TestListenerPortChanged
    (global_accelerator_test.go:157-343)."""

    def test_multiple_ports_changed(self):
        svc = svc_with_ports((8080, "TCP"), (9090, "TCP"))
        assert listener_port_changed_from_service(listener([80, 443]), svc)

    def test_ports_increased(self):
        svc = svc_with_ports((80, "TCP"), (443, "TCP"), (8443, "TCP"))
        assert listener_port_changed_from_service(listener([80, 443]), svc)

    def test_ports_decreased(self):
        svc = svc_with_ports((80, "TCP"))
        assert listener_port_changed_from_service(listener([80, 443]), svc)
