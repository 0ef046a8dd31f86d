"""Env-driven provider factory (composableresource_adapter.go:40-76 parity)."""

import pytest

from cro_amd.fabric.adapter import new_adapter
from cro_amd.fabric.mock import MockFabric


def test_invalid_device_resource_type(monkeypatch):
    monkeypatch.delenv("DEVICE_RESOURCE_TYPE", raising=False)
    with pytest.raises(ValueError, match="DEVICE_RESOURCE_TYPE"):
        new_adapter()
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "BOGUS")
    with pytest.raises(ValueError, match="DEVICE_RESOURCE_TYPE"):
        new_adapter()


def test_invalid_provider_type(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "NOPE")
    with pytest.raises(ValueError, match="CDI_PROVIDER_TYPE"):
        new_adapter()


def test_mock_provider(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "MOCK")
    adapter = new_adapter()
    assert adapter.device_resource_type == "DRA"
    assert isinstance(adapter.provider, MockFabric)


def test_injected_provider_short_circuits(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DEVICE_PLUGIN")
    fabric = MockFabric()
    adapter = new_adapter(provider=fabric)
    assert adapter.provider is fabric


def test_rke2_forbids_device_plugin(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DEVICE_PLUGIN")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "FTI_CDI")
    monkeypatch.delenv("FTI_CDI_CLUSTER_ID", raising=False)
    with pytest.raises(ValueError, match="RKE2"):
        new_adapter()


def test_fti_requires_api_type(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "FTI_CDI")
    monkeypatch.setenv("FTI_CDI_CLUSTER_ID", "cluster-1")
    monkeypatch.delenv("FTI_CDI_API_TYPE", raising=False)
    with pytest.raises(ValueError, match="FTI_CDI_API_TYPE"):
        new_adapter()


def test_fti_cm_and_fm_selected(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "FTI_CDI")
    monkeypatch.setenv("FTI_CDI_CLUSTER_ID", "cluster-1")
    monkeypatch.setenv("FTI_CDI_ENDPOINT", "fabric.example")
    monkeypatch.setenv("FTI_CDI_TENANT_ID", "tenant-1")
    monkeypatch.setenv("FTI_CDI_API_TYPE", "CM")
    assert new_adapter().provider.name == "fti-cm"
    monkeypatch.setenv("FTI_CDI_API_TYPE", "FM")
    assert new_adapter().provider.name == "fti-fm"


def test_sunfish_selected(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "SUNFISH")
    assert new_adapter().provider.name == "sunfish"


def test_nec_selected(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "NEC")
    monkeypatch.setenv("NEC_CDIM_IP", "10.0.0.1")
    monkeypatch.setenv("LAYOUT_APPLY_PORT", "8000")
    monkeypatch.setenv("CONFIGURATION_MANAGER_PORT", "8001")
    assert new_adapter().provider.name == "nec"


def test_fabric_tls_verify_knob(monkeypatch):
    monkeypatch.setenv("DEVICE_RESOURCE_TYPE", "DRA")
    monkeypatch.setenv("CDI_PROVIDER_TYPE", "FTI_CDI")
    monkeypatch.setenv("FTI_CDI_CLUSTER_ID", "cluster-1")
    monkeypatch.setenv("FTI_CDI_ENDPOINT", "fabric.example")
    monkeypatch.setenv("FTI_CDI_API_TYPE", "FM")
    assert new_adapter().provider.verify is True
    monkeypatch.setenv("CRO_FABRIC_TLS_VERIFY", "false")
    assert new_adapter().provider.verify is False
