"""Remaining fabric error personas from the reference's scenario matrix:
non-JSON bodies, REMOVE_FAILED reruns, unknown states, credential sourcing
failures (composableresource_controller_test.go Describe titles)."""

import httpx
import pytest

from cro_amd.fabric.base import FabricError, WaitingDeviceDetaching
from cro_amd.fabric.fti.cm import FTICMClient
from cro_amd.fabric.fti.fm import FTIFMClient
from cro_amd.fabric.fti.token import CachedToken, TokenError, file_credentials
from tests.conftest import make_resource
from tests.fakes import FakeFTIServer, make_jwt
from tests.test_fabric_fti import CREDS, MACHINE_UUID, seed_chain


def non_json_transport(token_ok=True):
    def handler(request: httpx.Request) -> httpx.Response:
        if "id_manager" in request.url.path:
            return httpx.Response(200, json={"access_token": make_jwt(), "token_type": "Bearer"})
        return httpx.Response(200, text="<html>this is not json</html>")

    return httpx.MockTransport(handler)


def test_cm_non_json_machine_body(client):
    seed_chain(client)
    transport = non_json_transport()
    c = FTICMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="c",
        token=CachedToken("f.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    with pytest.raises(FabricError, match="unmarshal"):
        c.add_resource(make_resource("gpu-1"))


def test_fm_non_json_machine_body(client):
    seed_chain(client)
    transport = non_json_transport()
    c = FTIFMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="cluster-1",
        token=CachedToken("f.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    r = make_resource("gpu-1")
    r.status.cdi_device_id = "res-x"
    with pytest.raises(FabricError, match="unmarshal"):
        c.remove_resource(r)


def test_fm_non_json_scaleup_response(client):
    seed_chain(client)

    def handler(request: httpx.Request) -> httpx.Response:
        if "id_manager" in request.url.path:
            return httpx.Response(200, json={"access_token": make_jwt(), "token_type": "Bearer"})
        if request.method == "PATCH":
            return httpx.Response(200, text="not json")
        return httpx.Response(404, json={"detail": {}})

    transport = httpx.MockTransport(handler)
    c = FTIFMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="cluster-1",
        token=CachedToken("f.example", credentials=CREDS, transport=transport),
        transport=transport,
    )
    with pytest.raises(FabricError, match="unmarshal"):
        c.add_resource(make_resource("gpu-1"))


def test_fm_attach_unknown_state(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_scaleup_response = server.fm_machine(
        resources=[server.fm_resource("GPU-odd", op_status="7")]
    )
    c = FTIFMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="cluster-1",
        token=CachedToken("f.example", credentials=CREDS, transport=server.transport()),
        transport=server.transport(),
    )
    with pytest.raises(FabricError, match="unknown state"):
        c.add_resource(make_resource("gpu-1"))


def test_fm_attach_device_not_in_response(client):
    server = FakeFTIServer()
    seed_chain(client)
    server.fm_scaleup_response = server.fm_machine(resources=[])
    c = FTIFMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="cluster-1",
        token=CachedToken("f.example", credentials=CREDS, transport=server.transport()),
        transport=server.transport(),
    )
    with pytest.raises(FabricError, match="can not find the added gpu"):
        c.add_resource(make_resource("gpu-1"))


def test_cm_remove_failed_reruns(client):
    """REMOVE_FAILED: the reason is surfaced into status.error and the
    scaledown is STILL issued (cm/client.go:206-215 semantics)."""
    server = FakeFTIServer()
    seed_chain(client)
    server.cm_machines[MACHINE_UUID] = server.cm_machine(
        devices=[
            FakeFTIServer.cm_device("GPU-x", status="REMOVE_FAILED", reason="stuck fabric port")
        ],
        device_count=1,
    )
    c = FTICMClient(
        client, endpoint="f.example", tenant_id="t", cluster_id="cluster-1",
        token=CachedToken("f.example", credentials=CREDS, transport=server.transport()),
        transport=server.transport(),
    )
    r = make_resource("gpu-1")
    r.status.device_id = "GPU-x"
    with pytest.raises(WaitingDeviceDetaching):
        c.remove_resource(r)
    assert r.status.error == "stuck fabric port"
    assert len(server.resize_calls) == 1  # scaledown still sent → rerun later


def test_file_credentials_missing(tmp_path):
    creds = file_credentials(str(tmp_path / "nope.json"))
    tok = CachedToken("f.example", credentials=creds)
    with pytest.raises(FileNotFoundError):
        tok.get_token()


def test_token_non_json_error_body():
    def handler(request):
        return httpx.Response(500, text="<html>oops</html>")

    tok = CachedToken("f.example", credentials=CREDS, transport=httpx.MockTransport(handler))
    with pytest.raises(TokenError, match="500"):
        tok.get_token()


def test_secret_dir_credentials(tmp_path):
    """Mounted-Secret credential source: per-key files, re-read per load
    (token.go:103-127 parity)."""
    from cro_amd.fabric.fti.token import secret_dir_credentials

    for key, val in [("username", "u"), ("password", "p"),
                     ("client_id", "cid"), ("client_secret", "cs"),
                     ("realm", "r1")]:
        (tmp_path / key).write_text(val + "\n")
    creds = secret_dir_credentials(str(tmp_path))
    assert creds() == {"username": "u", "password": "p", "client_id": "cid",
                       "client_secret": "cs", "realm": "r1"}
    # missing key → empty string, not a crash
    (tmp_path / "realm").unlink()
    assert creds()["realm"] == ""


def test_secret_rotation_picked_up_on_refresh(tmp_path):
    """A rotated Secret is used at the NEXT token refresh without restart:
    the credential fn re-reads the mount per fetch."""
    import time

    from cro_amd.fabric.fti.token import CachedToken, secret_dir_credentials
    from tests.fakes import FakeFTIServer

    for key, val in [("username", "old-user"), ("password", "p"),
                     ("client_id", "c"), ("client_secret", "s"),
                     ("realm", "r")]:
        (tmp_path / key).write_text(val)

    server = FakeFTIServer()
    server.token_exp = time.time() + 10  # inside leeway → refetch each call
    tok = CachedToken("fabric.example",
                      credentials=secret_dir_credentials(str(tmp_path)),
                      transport=server.transport())
    tok.get_token()
    assert server.last_token_request["username"] == "old-user"

    (tmp_path / "username").write_text("new-user")  # kubelet rotates the Secret
    tok.get_token()
    assert server.last_token_request["username"] == "new-user"


def test_default_credentials_resolution(tmp_path, monkeypatch):
    from cro_amd.fabric.fti.token import default_credentials

    monkeypatch.delenv("CRO_FTI_CREDENTIALS_DIR", raising=False)
    monkeypatch.delenv("CRO_FTI_CREDENTIALS_FILE", raising=False)
    monkeypatch.setenv("CRO_FTI_USERNAME", "env-user")
    assert default_credentials()()["username"] == "env-user"

    (tmp_path / "creds.json").write_text('{"username": "file-user"}')
    monkeypatch.setenv("CRO_FTI_CREDENTIALS_FILE", str(tmp_path / "creds.json"))
    assert default_credentials()()["username"] == "file-user"

    d = tmp_path / "secret"
    d.mkdir()
    (d / "username").write_text("dir-user")
    monkeypatch.setenv("CRO_FTI_CREDENTIALS_DIR", str(d))
    assert default_credentials()()["username"] == "dir-user"  # dir wins
