"""Upstream syncer: repairs fabric↔cluster drift.

Parity with upstreamsyncer_controller.go:49-165: every ``period`` the
fabric's ground truth (``get_resources``) is diffed against local
ComposableResource device IDs; a device attached upstream with no local CR is
tracked and, after ``grace_period``, a detach-CR is created carrying the
``ready-to-detach`` labels so the resource controller walks it straight
through Attaching→Online→Detaching bookkeeping and physically detaches it.
"""

from __future__ import annotations

import logging
import time
from typing import Dict

from ..api.v1alpha1.types import ComposableResource, ComposableResourceSpec
from ..fabric import Adapter
from ..nodeops.amdgpu import NodeOps
from ..runtime.client import Client
from ..utils import generate_composable_resource_name
from .composableresource import READY_TO_DETACH_CDI_LABEL, READY_TO_DETACH_LABEL

log = logging.getLogger(__name__)

DEFAULT_GRACE_PERIOD = 600.0  # 10 min (upstreamsyncer_controller.go:38)
DEFAULT_PERIOD = 60.0  # 1 min (:61)


class UpstreamSyncer:
    def __init__(
        self,
        client: Client,
        adapter: Adapter,
        node_ops: NodeOps,
        grace_period: float = DEFAULT_GRACE_PERIOD,
        recorder=None,
    ):
        from ..runtime.events import NullRecorder

        self.client = client
        self.adapter = adapter
        self.node_ops = node_ops
        self.grace_period = grace_period
        self.recorder = recorder or NullRecorder()
        self.missing_devices: Dict[str, float] = {}

    def sync(self) -> None:
        device_infos = self.adapter.provider.get_resources()

        existing_ids = set()
        for r in self.client.list(ComposableResource, copy=False):  # read-only
            if r.status.device_id:
                existing_ids.add(r.status.device_id)
            # a freshly created detach-CR carries the identity only in its
            # label until its first reconcile copies it into status — count
            # it or short sync periods create duplicate detach CRs for the
            # same orphan (the reference's 1-min period merely hides this
            # race, upstreamsyncer_controller.go:97-135)
            label_id = r.metadata.labels.get(READY_TO_DETACH_LABEL, "")
            if label_id:
                existing_ids.add(label_id)

        for info in device_infos:
            did = info.device_id
            if did in existing_ids:
                if did in self.missing_devices:
                    log.info("CR appeared for tracked device %s; untracking", did)
                    del self.missing_devices[did]
                continue
            first_seen = self.missing_devices.get(did)
            if first_seen is None:
                log.info("upstream device %s has no local CR; tracking with grace", did)
                self.missing_devices[did] = time.monotonic()
                self.recorder.warning(
                    ("Node", info.node_name),
                    "FabricDrift",
                    f"device {did} attached upstream with no local CR",
                )
            elif time.monotonic() - first_seen > self.grace_period:
                log.info("grace exceeded for %s; creating detach CR", did)
                try:
                    self._create_detach_cr(info)
                    del self.missing_devices[did]
                except Exception as exc:
                    log.error("failed to create detach CR for %s: %s", did, exc)

        upstream_ids = {d.device_id for d in device_infos}
        for tracked in list(self.missing_devices):
            if tracked not in upstream_ids:
                log.info("tracked device %s gone upstream; untracking", tracked)
                del self.missing_devices[tracked]

    def _create_detach_cr(self, info) -> None:
        self.node_ops.ensure_driver(info.node_name)
        cr = ComposableResource(
            spec=ComposableResourceSpec(
                type=info.device_type or "gpu",
                model=info.model,
                target_node=info.node_name,
                force_detach=False,
            )
        )
        cr.metadata.generateName = generate_composable_resource_name("gpu")
        cr.metadata.labels[READY_TO_DETACH_LABEL] = info.device_id
        cr.metadata.labels[READY_TO_DETACH_CDI_LABEL] = info.cdi_device_id
        self.client.create(cr)
        self.recorder.normal(
            ("Node", info.node_name),
            "DriftDetachCreated",
            f"detach CR created for unclaimed device {info.device_id}",
        )
