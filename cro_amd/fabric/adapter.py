"""Env-driven fabric provider factory.

Parity with NewComposableResourceAdapter
(internal/controller/composableresource_adapter.go:40-76):

* ``DEVICE_RESOURCE_TYPE`` ∈ {DEVICE_PLUGIN, DRA} — validated here;
* ``CDI_PROVIDER_TYPE`` ∈ {SUNFISH, NEC, FTI_CDI, MOCK} (MOCK is this build's
  in-process backend for tests/bench, the analog of the reference tests'
  httptest fabric);
* FTI_CDI further selects CM vs FM via ``FTI_CDI_API_TYPE``;
* an RKE2 cluster (no ``FTI_CDI_CLUSTER_ID``) cannot use DEVICE_PLUGIN.
"""

from __future__ import annotations

import os
from dataclasses import dataclass

from .base import FabricProvider

DEVICE_RESOURCE_TYPES = ("DEVICE_PLUGIN", "DRA")


@dataclass
class Adapter:
    device_resource_type: str
    provider: FabricProvider


def new_adapter(client=None, provider: FabricProvider = None) -> Adapter:
    device_resource_type = os.environ.get("DEVICE_RESOURCE_TYPE", "")
    if device_resource_type not in DEVICE_RESOURCE_TYPES:
        raise ValueError(
            f"the env variable DEVICE_RESOURCE_TYPE has an invalid value: '{device_resource_type}'"
        )

    if provider is not None:  # injected (tests, bench, embedded mock)
        return Adapter(device_resource_type, provider)

    # CRO_FABRIC_TLS_VERIFY=false disables certificate verification toward
    # the fabric manager (the reference tests' InsecureSkipVerify seam,
    # composableresource_controller_test.go:999-1005; production keeps it on)
    verify = os.environ.get("CRO_FABRIC_TLS_VERIFY", "true").lower() != "false"

    provider_type = os.environ.get("CDI_PROVIDER_TYPE", "")
    if provider_type == "MOCK":
        from .mock import MockFabric

        provider = MockFabric()
    elif provider_type == "SUNFISH":
        from .sunfish import SunfishClient

        provider = SunfishClient()
    elif provider_type == "NEC":
        from .nec import NECClient

        provider = NECClient(client)
    elif provider_type == "FTI_CDI":
        cluster_uuid = os.environ.get("FTI_CDI_CLUSTER_ID", "")
        if cluster_uuid == "" and device_resource_type == "DEVICE_PLUGIN":
            raise ValueError(
                "The cluster in RKE2 does not support DEVICE_PLUGIN, please use DRA"
            )
        api_type = os.environ.get("FTI_CDI_API_TYPE", "")
        if api_type == "CM":
            from .fti.cm import FTICMClient

            provider = FTICMClient(client, verify=verify)
        elif api_type == "FM":
            from .fti.fm import FTIFMClient

            provider = FTIFMClient(client, verify=verify)
        else:
            raise ValueError(
                f"the env variable FTI_CDI_API_TYPE has an invalid value: '{api_type}'"
            )
    else:
        raise ValueError(
            f"the env variable CDI_PROVIDER_TYPE has an invalid value: '{provider_type}'"
        )
    return Adapter(device_resource_type, provider)
