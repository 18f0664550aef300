"""Tolerance of real kube-apiserver wire payloads: objects coming from an
actual cluster carry many fields this framework doesn't model
(managedFields, clusterIP, conditions, …) — decoding must ignore them and
keep every field the controllers consume."""

from agac.apis import core as corev1
from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import from_dict, to_dict

# shaped like a real `kubectl get svc -o json` for an NLB service
REAL_SERVICE = {
    "apiVersion": "v1",
    "kind": "Service",
    "metadata": {
        "annotations": {
            "service.beta.kubernetes.io/aws-load-balancer-type": "external",
            "aws-global-accelerator-controller.h3poteto.dev/global-accelerator-managed": "true",
        },
        "creationTimestamp": "2026-01-12T08:33:21Z",
        "finalizers": ["service.kubernetes.io/load-balancer-cleanup"],
        "managedFields": [
            {
                "apiVersion": "v1",
                "fieldsType": "FieldsV1",
                "fieldsV1": {"f:metadata": {"f:annotations": {}}},
                "manager": "kubectl-client-side-apply",
                "operation": "Update",
                "time": "2026-01-12T08:33:21Z",
            }
        ],
        "name": "echoserver",
        "namespace": "default",
        "resourceVersion": "812345",
        "uid": "8d6b1d3c-88a5-43f6-9a0e-2a64e5a7e5bd",
    },
    "spec": {
        "allocateLoadBalancerNodePorts": True,
        "clusterIP": "10.100.23.45",
        "clusterIPs": ["10.100.23.45"],
        "externalTrafficPolicy": "Cluster",
        "internalTrafficPolicy": "Cluster",
        "ipFamilies": ["IPv4"],
        "ipFamilyPolicy": "SingleStack",
        "loadBalancerClass": "service.k8s.aws/nlb",
        "ports": [
            {
                "name": "http",
                "nodePort": 31380,
                "port": 80,
                "protocol": "TCP",
                "targetPort": 8080,
            }
        ],
        "selector": {"app": "echoserver"},
        "sessionAffinity": "None",
        "type": "LoadBalancer",
    },
    "status": {
        "loadBalancer": {
            "ingress": [
                {
                    "hostname": "k8s-default-echoserv-0123456789-aabbccddeeff0011.elb.ap-northeast-1.amazonaws.com"
                }
            ]
        }
    },
}

REAL_INGRESS = {
    "apiVersion": "networking.k8s.io/v1",
    "kind": "Ingress",
    "metadata": {
        "annotations": {
            "alb.ingress.kubernetes.io/listen-ports": '[{"HTTP": 80}, {"HTTPS": 443}]',
            "alb.ingress.kubernetes.io/scheme": "internet-facing",
            "kubernetes.io/ingress.class": "alb",
        },
        "generation": 2,
        "name": "echoserver",
        "namespace": "default",
        "resourceVersion": "812999",
        "uid": "0a1b2c3d-1111-2222-3333-444455556666",
    },
    "spec": {
        "rules": [
            {
                "host": "echo.example.com",
                "http": {
                    "paths": [
                        {
                            "backend": {
                                "service": {
                                    "name": "echoserver",
                                    "port": {"number": 8080},
                                }
                            },
                            "path": "/",
                            "pathType": "Prefix",
                        }
                    ]
                },
            }
        ]
    },
    "status": {
        "loadBalancer": {
            "ingress": [
                {"hostname": "k8s-default-echoserv-9876543210.ap-northeast-1.elb.amazonaws.com"}
            ]
        }
    },
}

REAL_STATUS_ERROR = {
    "kind": "Status",
    "apiVersion": "v1",
    "metadata": {},
    "status": "Failure",
    "message": 'services "nope" not found',
    "reason": "NotFound",
    "details": {"name": "nope", "kind": "services"},
    "code": 404,
}


def test_real_service_decodes():
    svc = from_dict(corev1.Service, REAL_SERVICE)
    assert svc.spec.type == "LoadBalancer"
    assert svc.spec.load_balancer_class == "service.k8s.aws/nlb"
    assert svc.spec.ports[0].port == 80
    assert svc.spec.ports[0].node_port == 31380
    assert svc.metadata.resource_version == "812345"
    assert svc.status.load_balancer.ingress[0].hostname.endswith("amazonaws.com")
    # the controllers' filters accept it
    from agac.controller.base import has_managed_annotation, was_load_balancer_service

    assert was_load_balancer_service(svc)
    assert has_managed_annotation(svc)
    # hostname parses (NLB form)
    from agac.cloudprovider.aws import get_lb_name_from_hostname

    name, region = get_lb_name_from_hostname(
        svc.status.load_balancer.ingress[0].hostname
    )
    # the aws-load-balancer-controller names NLBs k8s-<ns>-<svc>-<10 chars>;
    # only the DNS hash suffix is stripped
    assert name == "k8s-default-echoserv-0123456789"
    assert region == "ap-northeast-1"


def test_real_service_reencodes_known_fields_only():
    svc = from_dict(corev1.Service, REAL_SERVICE)
    out = to_dict(svc)
    assert "managedFields" not in out["metadata"]
    assert out["spec"]["ports"][0]["port"] == 80
    # update flows send back only modeled fields — a real apiserver accepts
    # partial objects on PUT only with full replacement, which is why the
    # production path should PATCH in future; documented limitation


def test_real_ingress_decodes():
    ingress = from_dict(corev1.Ingress, REAL_INGRESS)
    from agac.controller.base import was_alb_ingress
    from agac.cloudprovider.aws.global_accelerator import listener_for_ingress

    assert was_alb_ingress(ingress)
    ports, protocol = listener_for_ingress(ingress)
    assert ports == [80, 443]
    assert ingress.metadata.generation == 2


def test_real_status_error_maps_to_typed():
    from agac.kube import k8swire
    from agac.kube.store import NotFoundError

    err = k8swire.error_for_status(REAL_STATUS_ERROR, 404)
    assert isinstance(err, NotFoundError)
    assert "not found" in str(err)


def test_egb_from_kubectl_apply_shape():
    doc = {
        "apiVersion": "operator.h3poteto.dev/v1alpha1",
        "kind": "EndpointGroupBinding",
        "metadata": {
            "annotations": {
                "kubectl.kubernetes.io/last-applied-configuration": "{...}"
            },
            "name": "binding",
            "namespace": "default",
        },
        "spec": {
            "clientIPPreservation": False,
            "endpointGroupArn": "arn:aws:globalaccelerator::123:accelerator/x/listener/y/endpoint-group/z",
            "serviceRef": {"name": "echoserver"},
            "weight": 100,
        },
    }
    binding = from_dict(egb.EndpointGroupBinding, doc)
    assert binding.spec.weight == 100
    assert binding.spec.service_ref.name == "echoserver"


def test_owner_references_survive_update_roundtrip():
    """A real-cluster EGB with ownerReferences must not lose them when the
    controller PUTs back finalizer/spec changes (full-replacement update)."""
    doc = {
        "apiVersion": "operator.h3poteto.dev/v1alpha1",
        "kind": "EndpointGroupBinding",
        "metadata": {
            "name": "owned",
            "namespace": "default",
            "ownerReferences": [
                {
                    "apiVersion": "apps/v1",
                    "kind": "Deployment",
                    "name": "parent",
                    "uid": "u-1",
                    "controller": True,
                    "blockOwnerDeletion": True,
                }
            ],
        },
        "spec": {"endpointGroupArn": "arn:x"},
    }
    binding = from_dict(egb.EndpointGroupBinding, doc)
    assert binding.metadata.owner_references[0].name == "parent"
    assert binding.metadata.owner_references[0].controller is True
    out = to_dict(binding)
    ref = out["metadata"]["ownerReferences"][0]
    assert ref == {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "name": "parent",
        "uid": "u-1",
        "controller": True,
        "blockOwnerDeletion": True,
    }
