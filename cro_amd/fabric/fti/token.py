"""FTI OAuth2 password-grant token cache.

Protocol parity with internal/cdi/fti/token.go:103-175: Keycloak
password-grant against ``https://{endpoint}id_manager/realms/{realm}/
protocol/openid-connect/token``; expiry parsed from the JWT access-token
payload; double-checked cache with a 30 s expiry leeway (:58-101).

Credentials come from an injected provider.  The reference reads the
Kubernetes Secret ``composable-resource-operator-system/credentials``
(username/password/client_id/client_secret/realm); standalone deployments
use ``env_credentials`` (CRO_FTI_* variables) or ``file_credentials``.
"""

from __future__ import annotations

import base64
import binascii
import json
import os
import threading
import time
from typing import Callable, Dict, Optional

import httpx

TOKEN_REQUEST_TIMEOUT = 30.0
EXPIRY_LEEWAY = 30.0

CredentialsFn = Callable[[], Dict[str, str]]


class TokenError(Exception):
    pass


def env_credentials() -> Dict[str, str]:
    return {
        "username": os.environ.get("CRO_FTI_USERNAME", ""),
        "password": os.environ.get("CRO_FTI_PASSWORD", ""),
        "client_id": os.environ.get("CRO_FTI_CLIENT_ID", ""),
        "client_secret": os.environ.get("CRO_FTI_CLIENT_SECRET", ""),
        "realm": os.environ.get("CRO_FTI_REALM", ""),
    }


def file_credentials(path: str) -> CredentialsFn:
    def load() -> Dict[str, str]:
        with open(path) as f:
            return json.load(f)

    return load


SECRET_KEYS = ("username", "password", "client_id", "client_secret", "realm")


def secret_dir_credentials(path: str) -> CredentialsFn:
    """Kubernetes mounted-Secret credential source (token.go:103-127
    parity): the RBAC-pinned ``credentials`` Secret mounts as a directory
    of per-key files (username/password/client_id/client_secret/realm).
    Every load re-reads the files, so a rotated Secret is picked up at the
    next token refresh without a restart (kubelet atomically updates the
    mount's symlinks)."""

    def load() -> Dict[str, str]:
        creds: Dict[str, str] = {}
        for key in SECRET_KEYS:
            try:
                with open(os.path.join(path, key)) as f:
                    creds[key] = f.read().strip()
            except OSError:
                creds[key] = ""
        return creds

    return load


def default_credentials() -> CredentialsFn:
    """Production credential resolution, re-checked per token fetch:

    1. ``CRO_FTI_CREDENTIALS_DIR``  — mounted k8s Secret (rotation-aware);
       skipped while the optional mount is absent/empty, so the manifest
       can set the path unconditionally;
    2. ``CRO_FTI_CREDENTIALS_FILE`` — JSON file;
    3. ``CRO_FTI_*`` env variables.
    """

    def load() -> Dict[str, str]:
        cred_dir = os.environ.get("CRO_FTI_CREDENTIALS_DIR", "")
        if cred_dir and any(
            os.path.exists(os.path.join(cred_dir, k)) for k in SECRET_KEYS
        ):
            return secret_dir_credentials(cred_dir)()
        cred_file = os.environ.get("CRO_FTI_CREDENTIALS_FILE", "")
        if cred_file and os.path.exists(cred_file):
            return file_credentials(cred_file)()
        return env_credentials()

    return load


def parse_jwt_expiry(access_token: str) -> float:
    parts = access_token.split(".")
    if len(parts) != 3:
        raise TokenError(f"invalid access token: {access_token}")
    try:
        payload = base64.urlsafe_b64decode(parts[1] + "=" * (-len(parts[1]) % 4))
        return float(json.loads(payload)["exp"])
    except (binascii.Error, ValueError, KeyError) as exc:
        raise TokenError(f"failed to decode id_manager payload: {exc}") from exc


class CachedToken:
    def __init__(
        self,
        endpoint: str,
        credentials: Optional[CredentialsFn] = None,
        transport: Optional[httpx.BaseTransport] = None,
        verify: bool = True,
    ):
        if not endpoint.endswith("/"):
            endpoint += "/"
        self.endpoint = endpoint
        self.credentials = credentials or default_credentials()
        self.transport = transport
        self.verify = verify
        self.leeway = EXPIRY_LEEWAY
        self._lock = threading.Lock()
        # (token, expiry) kept as ONE tuple so the unlocked fast-path read
        # is atomic — two separate attribute reads could pair an old token
        # with a refreshed expiry under a concurrent refresh
        self._cached: "tuple[Optional[str], float]" = (None, 0.0)

    @property
    def _token(self) -> Optional[str]:
        return self._cached[0]

    @property
    def _expiry(self) -> float:
        return self._cached[1]

    def get_token(self) -> str:
        now = time.time()
        token, expiry = self._cached
        if token is not None and expiry - self.leeway > now:
            return token
        with self._lock:
            token, expiry = self._cached
            if token is not None and expiry - self.leeway > now:
                return token
            self._cached = self._fetch()
            return self._cached[0]

    def _fetch(self) -> "tuple[str, float]":
        creds = self.credentials()
        realm = creds.get("realm", "")
        url = f"https://{self.endpoint}id_manager/realms/{realm}/protocol/openid-connect/token"
        data = {
            "client_id": creds.get("client_id", ""),
            "client_secret": creds.get("client_secret", ""),
            "username": creds.get("username", ""),
            "password": creds.get("password", ""),
            "scope": "openid",
            "response_type": "id_token token",
            "grant_type": "password",
        }
        with httpx.Client(
            transport=self.transport, verify=self.verify, timeout=TOKEN_REQUEST_TIMEOUT
        ) as client:
            resp = client.post(url, data=data)
        if resp.status_code != 200:
            raise TokenError(
                f"http returned code: {resp.status_code}, response body: {resp.text}"
            )
        try:
            body = resp.json()
        except ValueError as exc:
            raise TokenError(f"failed to read id_manager response body into Token: {exc}")
        access_token = body.get("access_token", "")
        expiry = parse_jwt_expiry(access_token)
        return access_token, expiry
