"""Prometheus metrics.

The reference registers no custom metrics; BASELINE.json's
north star requires an attach→CDI-ready latency histogram plus reconcile
counters.  Buckets are sized for a composable fabric whose attach path is
dominated by fabric-manager RTT + PCIe rescan + amdgpu bind: sub-second with a
mock fabric, tens of seconds worst-case on real hardware.
"""

from __future__ import annotations

from prometheus_client import Counter, Gauge, Histogram, REGISTRY

_BUCKETS = (
    0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5,
    1.0, 2.5, 5.0, 10.0, 30.0, 60.0, 120.0, 300.0,
)


class Metrics:
    """Per-manager metric set (label ``controller`` distinguishes loops).

    prometheus_client registries are process-global; multiple Manager
    instances in one process (tests, bench ranks) share collectors via the
    class-level cache instead of re-registering.
    """

    _singleton = None

    def __new__(cls):
        if cls._singleton is None:
            cls._singleton = super().__new__(cls)
            cls._singleton._init_collectors()
        return cls._singleton

    def _init_collectors(self) -> None:
        self.attach_to_ready_seconds = Histogram(
            "cro_attach_to_ready_seconds",
            "Latency from ComposableResource entering Attaching to Online "
            "status write (CDI spec emitted and device visible)",
            buckets=_BUCKETS,
        )
        self.detach_seconds = Histogram(
            "cro_detach_seconds",
            "Latency from Detaching entry to device removal completion",
            buckets=_BUCKETS,
        )
        self.attach_phase_seconds = Histogram(
            "cro_attach_phase_seconds",
            "Per-phase attach timing (driver gate, fabric RTT, node refresh, "
            "CDI write, health probe) — the reference has no per-phase "
            "tracing (SURVEY.md §5.1)",
            ["phase"],
            buckets=_BUCKETS,
        )
        self.reconcile_total = Counter(
            "cro_reconcile_total",
            "Reconcile invocations by controller and outcome",
            ["controller", "result"],
        )
        self.fabric_request_seconds = Histogram(
            "cro_fabric_request_seconds",
            "Fabric-manager API round-trip time by operation",
            ["provider", "operation"],
            buckets=_BUCKETS,
        )
        self.devices_online = Gauge(
            "cro_devices_online", "ComposableResources currently Online"
        )

    # test helper: drop collectors so a fresh interpreter state can be faked
    @classmethod
    def _reset_for_tests(cls) -> None:
        if cls._singleton is not None:
            for collector in (
                cls._singleton.attach_to_ready_seconds,
                cls._singleton.detach_seconds,
                cls._singleton.attach_phase_seconds,
                cls._singleton.reconcile_total,
                cls._singleton.fabric_request_seconds,
                cls._singleton.devices_online,
            ):
                try:
                    REGISTRY.unregister(collector)
                except KeyError:
                    pass
            cls._singleton = None
