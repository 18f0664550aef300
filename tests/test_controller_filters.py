"""Controller notification filter tables (reference ga/service.go:18-26,
ga/ingress.go:19-27, ga/controller.go:243-259, r53/controller.go:243-252)."""

from agac.apis import core as corev1
from agac.apis.meta import ObjectMeta
from agac.controller.base import (
    has_hostname_annotation,
    has_managed_annotation,
    hostname_annotation_changed,
    managed_annotation_changed,
    objects_equal,
    was_alb_ingress,
    was_load_balancer_service,
)

MANAGED = "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed"
HOSTNAME = "aws-global-accelerator-controller.h3poteto.dev/route53-hostname"
LB_TYPE = "service.beta.kubernetes.io/aws-load-balancer-type"


def svc(svc_type="LoadBalancer", annotations=None, lb_class=None):
    return corev1.Service(
        metadata=ObjectMeta(name="s", namespace="d", annotations=annotations or {}),
        spec=corev1.ServiceSpec(type=svc_type, load_balancer_class=lb_class),
    )


def ingress(class_name=None, annotations=None):
    return corev1.Ingress(
        metadata=ObjectMeta(name="i", namespace="d", annotations=annotations or {}),
        spec=corev1.IngressSpec(ingress_class_name=class_name),
    )


class TestWasLoadBalancerService:
    def test_lb_type_annotation(self):
        assert was_load_balancer_service(svc(annotations={LB_TYPE: "nlb"}))

    def test_load_balancer_class(self):
        assert was_load_balancer_service(svc(lb_class="service.k8s.aws/nlb"))

    def test_plain_lb_service_rejected(self):
        # type LoadBalancer alone is NOT enough (cloud-provider-managed LBs)
        assert not was_load_balancer_service(svc())

    def test_cluster_ip_rejected(self):
        assert not was_load_balancer_service(
            svc(svc_type="ClusterIP", annotations={LB_TYPE: "nlb"})
        )


class TestWasALBIngress:
    def test_class_name_alb(self):
        assert was_alb_ingress(ingress(class_name="alb"))

    def test_other_class_name(self):
        assert not was_alb_ingress(ingress(class_name="nginx"))

    def test_legacy_annotation(self):
        # the reference accepts ANY kubernetes.io/ingress.class value
        assert was_alb_ingress(
            ingress(annotations={"kubernetes.io/ingress.class": "alb"})
        )
        assert was_alb_ingress(
            ingress(annotations={"kubernetes.io/ingress.class": "nginx"})
        )

    def test_no_class(self):
        assert not was_alb_ingress(ingress())


class TestAnnotationPredicates:
    def test_has_managed(self):
        assert has_managed_annotation(svc(annotations={MANAGED: "true"}))
        # presence, not value (reference checks `_, ok :=` only)
        assert has_managed_annotation(svc(annotations={MANAGED: ""}))
        assert not has_managed_annotation(svc())

    def test_managed_changed(self):
        with_it = svc(annotations={MANAGED: "true"})
        without = svc()
        assert managed_annotation_changed(without, with_it)
        assert managed_annotation_changed(with_it, without)
        assert not managed_annotation_changed(with_it, with_it)

    def test_hostname_predicates(self):
        with_it = svc(annotations={HOSTNAME: "a.example.com"})
        without = svc()
        assert has_hostname_annotation(with_it)
        assert not has_hostname_annotation(without)
        assert hostname_annotation_changed(without, with_it)
        assert not hostname_annotation_changed(with_it, with_it)


class TestObjectsEqual:
    def test_same_rv_short_circuits(self):
        a = svc()
        a.metadata.resource_version = "5"
        b = svc(annotations={"x": "y"})  # different content...
        b.metadata.resource_version = "5"  # ...but informers never do this
        assert objects_equal(a, b)

    def test_different_rv_compares_content(self):
        a = svc()
        a.metadata.resource_version = "5"
        b = svc()
        b.metadata.resource_version = "6"
        assert not objects_equal(a, b)  # rv is part of the wire content

    def test_no_rv_compares_content(self):
        assert objects_equal(svc(), svc())
        assert not objects_equal(svc(), svc(annotations={"x": "y"}))
