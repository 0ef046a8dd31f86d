"""Node → fabric machine-UUID resolution.

Parity with the reference's two paths:

* OpenShift chain (fti/cm/client.go:363-401, fti/fm/client.go:416-446):
  Node annotation ``machine.openshift.io/machine`` → Machine annotation
  ``metal3.io/BareMetalHost`` → BareMetalHost annotation
  ``cluster-manager.cdi.io/machine`` = the machine UUID.
* RKE2 (fti/fm/client.go:448-462): Node ``spec.providerID`` prefixed
  ``fsas-cdi://``.

Namespaced Machine/BMH objects are stored under "namespace/name" keys in
the cluster-scoped store.
"""

from __future__ import annotations

from ...api.v1alpha1.types import BareMetalHost, Machine, Node
from ...runtime.client import Client

MACHINE_ANNOTATION = "machine.openshift.io/machine"
BMH_ANNOTATION = "metal3.io/BareMetalHost"
CM_MACHINE_ANNOTATION = "cluster-manager.cdi.io/machine"
RKE2_PROVIDER_PREFIX = "fsas-cdi://"


class MachineResolutionError(Exception):
    pass


def resolve_machine_id_openshift(client: Client, node_name: str) -> str:
    node = client.get(Node, node_name)
    machine_info = node.metadata.annotations.get(MACHINE_ANNOTATION, "")
    parts = machine_info.split("/")
    if len(parts) != 2:
        raise MachineResolutionError(
            f"failed to get annotation '{MACHINE_ANNOTATION}' from Node "
            f"{node_name}, now is '{machine_info}'"
        )
    machine = client.get(Machine, machine_info)
    bmh_info = machine.metadata.annotations.get(BMH_ANNOTATION, "")
    bmh_parts = bmh_info.split("/")
    if len(bmh_parts) != 2:
        raise MachineResolutionError(
            f"failed to get annotation '{BMH_ANNOTATION}' from Machine "
            f"{machine.metadata.name}, now is '{bmh_info}'"
        )
    bmh = client.get(BareMetalHost, bmh_info)
    machine_uuid = bmh.metadata.annotations.get(CM_MACHINE_ANNOTATION, "")
    if not machine_uuid:
        raise MachineResolutionError(
            f"failed to get annotation '{CM_MACHINE_ANNOTATION}' from "
            f"BareMetalHost {bmh.metadata.name}, now is '{machine_uuid}'"
        )
    return machine_uuid


def resolve_machine_id_rke2(client: Client, node_name: str) -> str:
    node = client.get(Node, node_name)
    provider_id = node.status.provider_id
    if not provider_id.startswith(RKE2_PROVIDER_PREFIX):
        raise MachineResolutionError(
            f"invalid format: expected 'fsas-cdi://machineUUID', now is '{provider_id}'"
        )
    return provider_id[len(RKE2_PROVIDER_PREFIX):]


def resolve_machine_id(client: Client, node_name: str, cluster_id: str) -> str:
    """FM rule (fm/client.go:416-464): OpenShift chain when a cluster id is
    configured, else the RKE2 providerID."""
    if cluster_id:
        return resolve_machine_id_openshift(client, node_name)
    return resolve_machine_id_rke2(client, node_name)
