"""CRD manifest generation (controller-gen analog).

Emits the two CustomResourceDefinitions with schema parity to the reference
(config/crd/bases/cro.hpsys.ibm.ie.com_composabilityrequests.yaml and
_composableresources.yaml): same group, cluster scope, enums, minimums and
status subresource.  ``python -m cro_amd.api.v1alpha1.crd [outdir]``
regenerates; tests assert the committed files match.
"""

from __future__ import annotations

import os
import sys

import yaml

from . import types as t

GROUP = t.GROUP
VERSION = t.VERSION


def _other_spec_schema() -> dict:
    return {
        "type": "object",
        "properties": {
            "milli_cpu": {"type": "integer", "format": "int64", "minimum": 0},
            "memory": {"type": "integer", "format": "int64", "minimum": 0},
            "ephemeral_storage": {"type": "integer", "format": "int64", "minimum": 0},
            "allowed_pod_number": {"type": "integer", "format": "int64", "minimum": 0},
        },
    }


def _scalar_resource_details_schema() -> dict:
    return {
        "type": "object",
        "required": ["type", "model", "size"],
        "properties": {
            "type": {"type": "string", "enum": ["gpu", "cxlmemory"]},
            "model": {"type": "string", "minLength": 1},
            "size": {"type": "integer", "format": "int64", "minimum": 0},
            "force_detach": {"type": "boolean"},
            "allocation_policy": {
                "type": "string",
                "enum": ["samenode", "differentnode"],
                "default": "samenode",
            },
            "target_node": {"type": "string"},
            "other_spec": _other_spec_schema(),
        },
    }


def composability_request_crd() -> dict:
    scalar_status = {
        "type": "object",
        "required": ["state"],
        "properties": {
            "state": {"type": "string"},
            "device_id": {"type": "string"},
            "cdi_device_id": {"type": "string"},
            "node_name": {"type": "string"},
            "error": {"type": "string"},
        },
    }
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"composabilityrequests.{GROUP}"},
        "spec": {
            "group": GROUP,
            "names": {
                "kind": "ComposabilityRequest",
                "listKind": "ComposabilityRequestList",
                "plural": "composabilityrequests",
                "singular": "composabilityrequest",
            },
            "scope": "Cluster",
            "versions": [
                {
                    "name": VERSION,
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "apiVersion": {"type": "string"},
                                "kind": {"type": "string"},
                                "metadata": {"type": "object"},
                                "spec": {
                                    "type": "object",
                                    "required": ["resource"],
                                    "properties": {
                                        "resource": _scalar_resource_details_schema()
                                    },
                                },
                                "status": {
                                    "type": "object",
                                    "required": ["state"],
                                    "properties": {
                                        "state": {"type": "string"},
                                        "error": {"type": "string"},
                                        "resources": {
                                            "type": "object",
                                            "additionalProperties": scalar_status,
                                        },
                                        "scalarResource": _scalar_resource_details_schema(),
                                    },
                                },
                            },
                        }
                    },
                }
            ],
        },
    }


def composable_resource_crd() -> dict:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"composableresources.{GROUP}"},
        "spec": {
            "group": GROUP,
            "names": {
                "kind": "ComposableResource",
                "listKind": "ComposableResourceList",
                "plural": "composableresources",
                "singular": "composableresource",
            },
            "scope": "Cluster",
            "versions": [
                {
                    "name": VERSION,
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                    "schema": {
                        "openAPIV3Schema": {
                            "type": "object",
                            "properties": {
                                "apiVersion": {"type": "string"},
                                "kind": {"type": "string"},
                                "metadata": {"type": "object"},
                                "spec": {
                                    "type": "object",
                                    "required": ["type", "model", "target_node"],
                                    "properties": {
                                        "type": {
                                            "type": "string",
                                            "enum": ["gpu", "cxlmemory"],
                                        },
                                        "model": {"type": "string"},
                                        "target_node": {"type": "string"},
                                        "force_detach": {"type": "boolean"},
                                    },
                                },
                                "status": {
                                    "type": "object",
                                    "required": ["state"],
                                    "properties": {
                                        "state": {"type": "string"},
                                        "error": {"type": "string"},
                                        "device_id": {"type": "string"},
                                        "cdi_device_id": {"type": "string"},
                                        # extension: resumable async-fabric
                                        # wait (types.py ComposableResourceStatus)
                                        "fabric_wait_started": {"type": "string"},
                                    },
                                },
                            },
                        }
                    },
                }
            ],
        },
    }


def write_crds(outdir: str) -> list:
    os.makedirs(outdir, exist_ok=True)
    written = []
    for name, crd in (
        (f"cro.hpsys.ibm.ie.com_composabilityrequests.yaml", composability_request_crd()),
        (f"cro.hpsys.ibm.ie.com_composableresources.yaml", composable_resource_crd()),
    ):
        path = os.path.join(outdir, name)
        with open(path, "w") as f:
            yaml.safe_dump(crd, f, sort_keys=False)
        written.append(path)
    return written


if __name__ == "__main__":
    outdir = sys.argv[1] if len(sys.argv) > 1 else "config/crd/bases"
    for path in write_crds(outdir):
        print(path)
