"""Typed API objects for the cro.hpsys.ibm.ie.com/v1alpha1 group.

Field-for-field schema parity with the reference CRDs
(/root/reference/api/v1alpha1/composabilityrequest_types.go:36-80 and
/root/reference/api/v1alpha1/composableresource_types.go:27-41): same JSON
field names (``force_detach``, ``allocation_policy``, ``other_spec``,
``scalarResource``, ...), same enums, cluster scope, and a separate status
subresource (enforced by the store, see cro_amd/runtime/store.py).

Also defines the minimal cluster objects the operator consumes (Node,
ResourceSlice, DeviceTaintRule) — enough surface for the reconcile semantics
without depending on a kubernetes client package.
"""

from __future__ import annotations

from typing import ClassVar, Dict, List, Optional

from pydantic import BaseModel, ConfigDict, Field

GROUP = "cro.hpsys.ibm.ie.com"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"

# ComposableResource / ComposabilityRequest state constants.  The reference
# declares one set (composabilityrequest_types.go:23-30) but its controllers
# use literal strings; we name the literal-string machine states directly.
REQUEST_STATES = ("", "NodeAllocating", "Updating", "Running", "Cleaning", "Deleting")
RESOURCE_STATES = ("", "Attaching", "Online", "Detaching", "Deleting")


class _Model(BaseModel):
    model_config = ConfigDict(populate_by_name=True, extra="forbid")

    def clone(self):
        """Independent deep copy via a dump/validate roundtrip — ~2.5×
        faster than ``model_copy(deep=True)`` (copy.deepcopy underneath).
        The store copies on every get/list/write/watch fan-out, making
        this the single largest plumbing cost under the GIL (measured:
        51 µs vs 20 µs for an 8-device request)."""
        return self.__class__.model_validate(self.model_dump(by_alias=True))


class ObjectMeta(_Model):
    """Subset of k8s ObjectMeta the operator relies on (cluster-scoped)."""

    name: str = ""
    generateName: str = ""
    uid: str = ""
    resourceVersion: str = ""
    generation: int = 0
    creationTimestamp: Optional[str] = None
    deletionTimestamp: Optional[str] = None
    labels: Dict[str, str] = Field(default_factory=dict)
    annotations: Dict[str, str] = Field(default_factory=dict)
    finalizers: List[str] = Field(default_factory=list)


class K8sObject(_Model):
    """Base for all stored objects: apiVersion/kind/metadata."""

    KIND: ClassVar[str] = ""
    apiVersion: str = API_VERSION
    kind: str = ""
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)

    def model_post_init(self, __context) -> None:
        if not self.kind:
            self.kind = self.KIND

    @property
    def name(self) -> str:
        return self.metadata.name


# ---------------------------------------------------------------------------
# ComposabilityRequest (user-facing fleet request)
# ---------------------------------------------------------------------------


class NodeSpecRequirements(_Model):
    """``other_spec`` capacity requirements.

    Parity: NodeSpec, composabilityrequest_types.go:56-64.
    """

    milli_cpu: int = 0
    memory: int = 0
    ephemeral_storage: int = 0
    allowed_pod_number: int = 0


class ScalarResourceDetails(_Model):
    """Parity: ScalarResourceDetails, composabilityrequest_types.go:40-54."""

    type: str  # "gpu" | "cxlmemory"
    model: str
    size: int
    force_detach: bool = False
    allocation_policy: str = "samenode"  # "samenode" | "differentnode"
    target_node: str = ""
    other_spec: Optional[NodeSpecRequirements] = None


class ScalarResourceStatus(_Model):
    """Parity: ScalarResourceStatus, composabilityrequest_types.go:73-79."""

    state: str = ""
    device_id: str = ""
    cdi_device_id: str = ""
    node_name: str = ""
    error: str = ""


class ComposabilityRequestSpec(_Model):
    resource: ScalarResourceDetails


class ComposabilityRequestStatus(_Model):
    """Parity: ComposabilityRequestStatus, composabilityrequest_types.go:67-71."""

    state: str = ""
    error: str = ""
    resources: Dict[str, ScalarResourceStatus] = Field(default_factory=dict)
    scalarResource: Optional[ScalarResourceDetails] = None


class ComposabilityRequest(K8sObject):
    KIND: ClassVar[str] = "ComposabilityRequest"
    spec: Optional[ComposabilityRequestSpec] = None
    status: ComposabilityRequestStatus = Field(default_factory=ComposabilityRequestStatus)


# ---------------------------------------------------------------------------
# ComposableResource (internal per-device object)
# ---------------------------------------------------------------------------


class ComposableResourceSpec(_Model):
    """Parity: ComposableResourceSpec, composableresource_types.go:27-33."""

    type: str
    model: str
    target_node: str
    force_detach: bool = False


class ComposableResourceStatus(_Model):
    """Parity: ComposableResourceStatus, composableresource_types.go:36-41.

    ``fabric_wait_started`` is an extension beyond the reference: the
    RFC3339 start of an in-progress async fabric wait, persisted so a
    restarted operator resumes the exponential poll at the max interval
    instead of hammering the fabric from the base interval again (the
    reference keeps no such state and re-polls at its fixed 30 s quantum).
    """

    state: str = ""
    error: str = ""
    device_id: str = ""
    cdi_device_id: str = ""
    fabric_wait_started: str = ""


class ComposableResource(K8sObject):
    KIND: ClassVar[str] = "ComposableResource"
    spec: Optional[ComposableResourceSpec] = None
    status: ComposableResourceStatus = Field(default_factory=ComposableResourceStatus)


# ---------------------------------------------------------------------------
# Cluster objects the operator consumes (minimal faithful surface)
# ---------------------------------------------------------------------------


class NodeCapacity(_Model):
    """Node Status.Capacity figures compared in other_spec admission
    (nodes.go:84-117 uses Capacity, not Allocatable)."""

    milli_cpu: int = 0
    memory: int = 0
    ephemeral_storage: int = 0
    allowed_pod_number: int = 0


class NodeStatus(_Model):
    capacity: NodeCapacity = Field(default_factory=NodeCapacity)
    # amdgpu enumeration surfaced by the node agent (device-id -> render node)
    provider_id: str = ""


class Node(K8sObject):
    KIND: ClassVar[str] = "Node"
    apiVersion: str = "v1"
    status: NodeStatus = Field(default_factory=NodeStatus)


class ResourceSliceDevice(_Model):
    """One device row of a DRA ResourceSlice (uuid attribute is the seam the
    reference matches visibility on, gpus.go:207-239)."""

    name: str = ""
    uuid: str = ""
    model: str = ""
    node: str = ""
    attributes: Dict[str, str] = Field(default_factory=dict)


class ResourceSliceSpec(_Model):
    node_name: str = ""
    driver: str = "gpu.amd.com"
    pool: str = ""
    devices: List[ResourceSliceDevice] = Field(default_factory=list)


class ResourceSlice(K8sObject):
    KIND: ClassVar[str] = "ResourceSlice"
    apiVersion: str = "resource.k8s.io/v1alpha3"
    spec: ResourceSliceSpec = Field(default_factory=ResourceSliceSpec)


class DeviceTaintRuleSpec(_Model):
    """DRA DeviceTaintRule keyed by device uuid (gpus.go:894-989): NoSchedule
    taint applied to a composed device while it is being detached."""

    device_uuid: str = ""
    driver: str = "gpu.amd.com"
    effect: str = "NoSchedule"
    reason: str = ""


class DeviceTaintRule(K8sObject):
    KIND: ClassVar[str] = "DeviceTaintRule"
    apiVersion: str = "resource.k8s.io/v1alpha3"
    spec: DeviceTaintRuleSpec = Field(default_factory=DeviceTaintRuleSpec)


class DaemonSetSpec(_Model):
    """Minimal DaemonSet surface for the rolling-restart path: the template
    annotations carry ``kubectl.kubernetes.io/restartedAt``
    (nodes.go:35-76)."""

    template_annotations: Dict[str, str] = Field(default_factory=dict)


class DaemonSetStatus(_Model):
    desired_number_scheduled: int = 0
    number_ready: int = 0
    current_number_scheduled: int = 0
    number_unavailable: int = 0
    number_misscheduled: int = 0


class DaemonSet(K8sObject):
    KIND: ClassVar[str] = "DaemonSet"
    apiVersion: str = "apps/v1"
    spec: DaemonSetSpec = Field(default_factory=DaemonSetSpec)
    status: DaemonSetStatus = Field(default_factory=DaemonSetStatus)


class Machine(K8sObject):
    """OpenShift Machine (annotation carrier in the FTI node→machine
    chain). Registered like the reference registers the machine scheme
    (cmd/main.go:52-59); stored under "namespace/name" keys."""

    KIND: ClassVar[str] = "Machine"
    apiVersion: str = "machine.openshift.io/v1beta1"


class BareMetalHost(K8sObject):
    """metal3 BareMetalHost (carries cluster-manager.cdi.io/machine)."""

    KIND: ClassVar[str] = "BareMetalHost"
    apiVersion: str = "metal3.io/v1alpha1"


class DeviceConfigDriver(_Model):
    enable: bool = False
    daemonset_name: str = "amd-gpu-driver"
    # rootfs of the driver container on the host — module ops chroot here
    # in container mode (the /run/nvidia/driver analog, gpus.go:566-749)
    driver_root: str = "/run/amdgpu-driver"


class DeviceConfigSpec(_Model):
    driver: DeviceConfigDriver = Field(default_factory=DeviceConfigDriver)


class DeviceConfig(K8sObject):
    """Minimal AMD GPU operator DeviceConfig surface — the ClusterPolicy
    analog in the reference's driver-detection chain (gpus.go:97-127:
    driver daemonset pod → Container; ClusterPolicy driver enabled →
    Container; else modinfo → Host). Only the containerized-driver gate is
    modeled; the full CRD belongs to the AMD GPU operator."""

    KIND: ClassVar[str] = "DeviceConfig"
    apiVersion: str = "amd.com/v1alpha1"
    spec: DeviceConfigSpec = Field(default_factory=DeviceConfigSpec)


class Event(K8sObject):
    """Lifecycle event stream (beyond the reference, which emits no
    Kubernetes Events anywhere — no EventRecorder in any controller).
    Flat core/v1-Event-shaped record: who it is about, what happened, and
    a dedup count bumped on repeats. In cluster mode these map 1:1 onto
    corev1 Events; in standalone mode they are served by the REST API and
    read with ``croctl events``."""

    KIND: ClassVar[str] = "Event"
    apiVersion: str = "v1"
    involved_kind: str = ""
    involved_name: str = ""
    type: str = "Normal"  # Normal | Warning
    reason: str = ""
    message: str = ""
    count: int = 1
    first_seen: str = ""
    last_seen: str = ""
    source: str = "cro-amd"


class LeaseSpec(_Model):
    """coordination.k8s.io/v1 LeaseSpec field-for-field."""

    holderIdentity: str = ""
    leaseDurationSeconds: int = 15
    acquireTime: Optional[str] = None  # RFC3339 micro time
    renewTime: Optional[str] = None
    leaseTransitions: int = 0


class Lease(K8sObject):
    """coordination.k8s.io/v1 Lease — leader election (parity with the
    reference manager's lease-based election, cmd/main.go:137-155:
    LeaderElectionID c5744f42.hpsys.ibm.ie.com). Held/renewed by
    runtime.lease.LeaderElector through the normal client surface, so the
    same election works over the embedded store or the remote API."""

    KIND: ClassVar[str] = "Lease"
    apiVersion: str = "coordination.k8s.io/v1"
    spec: LeaseSpec = Field(default_factory=LeaseSpec)


ALL_KINDS = {
    cls.KIND: cls
    for cls in (
        ComposabilityRequest,
        ComposableResource,
        Node,
        ResourceSlice,
        DeviceTaintRule,
        DaemonSet,
        DeviceConfig,
        Event,
        Lease,
        Machine,
        BareMetalHost,
    )
}
