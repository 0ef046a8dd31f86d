"""cro_amd — MI355X-native composable-resource operator.

A from-scratch, AMD Instinct MI355X (gfx950) native implementation of the
capabilities of CoHDI/composable-resource-operator: hot-attach / hot-detach of
composable GPUs over a CXL/PCIe fabric, driven by a reconcile loop over the
``ComposabilityRequest`` / ``ComposableResource`` API (schema parity with
/root/reference/api/v1alpha1/*_types.go), with an amdgpu/ROCm/KFD device path,
Container Device Interface (CDI) spec emission, a validating admission layer,
an upstream drift syncer, and Prometheus attach-latency metrics.

Layout (mirrors SURVEY.md §1 layer map):
  api/        CRD-equivalent typed objects (pydantic), schema generation
  runtime/    controller-runtime equivalent: store, client, workqueue,
              controller, manager
  fabric/     composable-fabric manager clients ("CDI providers"):
              FTI CM / FTI FM / NEC CDIM / Sunfish + in-process mock
  nodeops/    amdgpu node path: KFD topology, visibility, loads, drain,
              CDI spec writer, GPU health probe (HIP, gfx950)
  controllers/ the three reconcilers (request / resource / upstream syncer)
  webhook/    admission validation rules + HTTP admission server
"""

__version__ = "0.1.0"

GROUP = "cro.hpsys.ibm.ie.com"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"
