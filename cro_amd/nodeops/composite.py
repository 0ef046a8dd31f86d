"""CompositeNodeOps: per-resource-type dispatch over NodeOps backends.

The CRD's ``type`` enum is {gpu, cxlmemory}; the device lifecycles differ
(KFD/CDI/probe vs CXL memdev/dax), so the controller resolves the backend
per resource via ``node_ops.for_type(resource.spec.type)``.
"""

from __future__ import annotations

from typing import Dict

from .amdgpu import NodeOps


class CompositeNodeOps:
    """Deliberately NOT a NodeOps subclass: the base class defines the verb
    methods (as NotImplementedError stubs), which would shadow the
    ``__getattr__`` delegation below."""

    def __init__(self, backends: Dict[str, NodeOps]):
        if not backends:
            raise ValueError("CompositeNodeOps needs at least one backend")
        self.backends = backends
        self._default = backends.get("gpu") or next(iter(backends.values()))

    def for_type(self, resource_type: str) -> NodeOps:
        backend = self.backends.get(resource_type)
        if backend is None:
            raise KeyError(
                f"no node-ops backend for resource type {resource_type!r} "
                f"(have {sorted(self.backends)})"
            )
        return backend

    # direct calls (syncer's ensure_driver, harness helpers) hit the default
    def __getattr__(self, item):
        return getattr(self._default, item)
