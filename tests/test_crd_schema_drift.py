"""CRD manifest ↔ Python types drift check (the reference's manifests.yml
CI job regenerates with controller-gen and fails on diff; here the CRD yaml
is hand-written, so this test asserts it stays in sync with the dataclasses
and the store's validation)."""

import yaml

from agac.apis import endpointgroupbinding as egb
from agac.apis.meta import ObjectMeta, to_dict

CRD_PATH = "config/crd/endpointgroupbindings.yaml"


def load_crd():
    import os

    path = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), CRD_PATH)
    with open(path) as f:
        return yaml.safe_load(f)


def full_binding():
    return egb.EndpointGroupBinding(
        metadata=ObjectMeta(name="x", namespace="d"),
        spec=egb.EndpointGroupBindingSpec(
            endpoint_group_arn="arn:x",
            client_ip_preservation=True,
            weight=1,
            service_ref=egb.ServiceReference(name="s"),
            ingress_ref=egb.IngressReference(name="i"),
        ),
        status=egb.EndpointGroupBindingStatus(endpoint_ids=["a"], observed_generation=1),
    )


def test_group_version_names():
    crd = load_crd()
    assert crd["spec"]["group"] == egb.GROUP
    assert crd["spec"]["names"]["kind"] == egb.EndpointGroupBinding.kind
    assert crd["spec"]["names"]["plural"] == "endpointgroupbindings"
    (version,) = crd["spec"]["versions"]
    assert version["name"] == egb.VERSION
    assert version["subresources"] == {"status": {}}


def test_spec_properties_match_dataclass():
    crd = load_crd()
    (version,) = crd["spec"]["versions"]
    schema = version["schema"]["openAPIV3Schema"]["properties"]
    spec_props = set(schema["spec"]["properties"])
    serialized = set(to_dict(full_binding())["spec"])
    assert spec_props == serialized, (
        f"CRD spec fields {spec_props} != dataclass wire fields {serialized}"
    )
    assert schema["spec"]["required"] == ["endpointGroupArn"]


def test_status_properties_match_dataclass():
    crd = load_crd()
    (version,) = crd["spec"]["versions"]
    schema = version["schema"]["openAPIV3Schema"]["properties"]
    status_props = set(schema["status"]["properties"])
    serialized = set(to_dict(full_binding())["status"])
    assert status_props == serialized


def test_printer_columns_reference_real_paths():
    crd = load_crd()
    (version,) = crd["spec"]["versions"]
    body = to_dict(full_binding())
    body["metadata"]["creationTimestamp"] = "2026-01-01T00:00:00Z"
    for col in version["additionalPrinterColumns"]:
        path = col["jsonPath"].lstrip(".").split(".")
        node = body
        for part in path:
            assert part in node, f"printer column {col['jsonPath']} dangling at {part}"
            node = node[part]


def test_k8swire_registry_matches_crd():
    from agac.kube import k8swire

    crd = load_crd()
    gvr = k8swire.gvr_for_kind("EndpointGroupBinding")
    assert gvr.group == crd["spec"]["group"]
    assert gvr.plural == crd["spec"]["names"]["plural"]
    assert gvr.version == crd["spec"]["versions"][0]["name"]


def test_sample_manifest_validates_against_schema_and_store():
    """The shipped EGB sample must satisfy both the CRD yaml constraints
    and the store-side validation."""
    import os

    from agac.apis.meta import from_dict
    from agac.kube.client import InMemoryKubeClient

    path = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "config", "samples", "endpointgroupbinding.yaml",
    )
    with open(path) as f:
        doc = yaml.safe_load(f)
    binding = from_dict(egb.EndpointGroupBinding, doc)
    assert binding.spec.endpoint_group_arn.startswith("arn:aws:globalaccelerator")
    client = InMemoryKubeClient()
    created = client.create(binding)  # store validation passes
    assert created.metadata.generation == 1
