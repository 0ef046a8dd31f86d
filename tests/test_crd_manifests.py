"""CRD manifest generation: schema parity with the reference CRDs and
no-drift guarantee for the committed files."""

import os

import yaml

from cro_amd.api.v1alpha1.crd import (
    composability_request_crd,
    composable_resource_crd,
)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BASES = os.path.join(REPO, "config", "crd", "bases")


def test_request_crd_schema_parity():
    crd = composability_request_crd()
    assert crd["metadata"]["name"] == "composabilityrequests.cro.hpsys.ibm.ie.com"
    spec = crd["spec"]
    assert spec["scope"] == "Cluster"
    v = spec["versions"][0]
    assert v["name"] == "v1alpha1"
    assert v["subresources"] == {"status": {}}
    resource = v["schema"]["openAPIV3Schema"]["properties"]["spec"]["properties"]["resource"]
    assert resource["properties"]["type"]["enum"] == ["gpu", "cxlmemory"]
    assert resource["properties"]["allocation_policy"]["enum"] == ["samenode", "differentnode"]
    assert resource["properties"]["allocation_policy"]["default"] == "samenode"
    assert resource["properties"]["size"]["minimum"] == 0
    assert resource["properties"]["model"]["minLength"] == 1
    assert set(resource["required"]) == {"type", "model", "size"}
    other = resource["properties"]["other_spec"]["properties"]
    assert set(other) == {"milli_cpu", "memory", "ephemeral_storage", "allowed_pod_number"}
    status = v["schema"]["openAPIV3Schema"]["properties"]["status"]
    assert "scalarResource" in status["properties"]
    per_device = status["properties"]["resources"]["additionalProperties"]["properties"]
    assert set(per_device) == {"state", "device_id", "cdi_device_id", "node_name", "error"}


def test_resource_crd_schema_parity():
    crd = composable_resource_crd()
    spec_schema = crd["spec"]["versions"][0]["schema"]["openAPIV3Schema"]["properties"]["spec"]
    assert set(spec_schema["required"]) == {"type", "model", "target_node"}
    assert spec_schema["properties"]["type"]["enum"] == ["gpu", "cxlmemory"]
    status = crd["spec"]["versions"][0]["schema"]["openAPIV3Schema"]["properties"]["status"]
    # reference fields + the documented fabric_wait_started extension
    # (resumable async-fabric wait, types.py ComposableResourceStatus)
    assert set(status["properties"]) == {
        "state", "error", "device_id", "cdi_device_id", "fabric_wait_started"}


def test_committed_manifests_match_generator():
    for fname, gen in (
        ("cro.hpsys.ibm.ie.com_composabilityrequests.yaml", composability_request_crd),
        ("cro.hpsys.ibm.ie.com_composableresources.yaml", composable_resource_crd),
    ):
        with open(os.path.join(BASES, fname)) as f:
            committed = yaml.safe_load(f)
        assert committed == gen(), f"{fname} drifted; run `make manifests`"
