"""RemoteNodeExec: the NodeExec surface over per-node agent HTTP APIs.

The off-node half of the split deployment (the reference's SPDY pod-exec
analog, gpus.go:1040-1067): controllers resolve a node name to its agent
URL and drive the same read/write/run/list verbs the local implementation
provides — so AmdNodeOps, the KFD parser and the CDI writer run unchanged
against remote nodes.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple, Union

import httpx

from .execs import ExecError, NodeExec

# node name → agent base URL; a callable allows dynamic discovery (e.g.
# from DaemonSet pod IPs in cluster mode)
Resolver = Union[Dict[str, str], Callable[[str], str]]


class RemoteNodeExec(NodeExec):
    def __init__(
        self,
        resolver: Resolver,
        transport: Optional[httpx.BaseTransport] = None,
        token: Optional[str] = None,
    ):
        import os

        self._resolver = resolver
        if token is None:
            token = os.environ.get("CRO_AGENT_TOKEN", "")
        headers = {"Authorization": f"Bearer {token}"} if token else None
        self._http = httpx.Client(transport=transport, timeout=90, headers=headers)

    def _url(self, node: str, path: str) -> str:
        if callable(self._resolver):
            base = self._resolver(node)
        else:
            base = self._resolver.get(node, "")
        if not base:
            raise ExecError(f"no agent endpoint known for node {node!r}")
        return base.rstrip("/") + path

    def run(self, node: str, argv: List[str], timeout: float = 60.0) -> Tuple[int, str, str]:
        resp = self._http.post(
            self._url(node, "/agent/run"),
            json={"argv": argv, "timeout": timeout},
            timeout=timeout + 30,
        )
        if resp.status_code == 400:
            raise ExecError(resp.json().get("detail", resp.text))
        resp.raise_for_status()
        body = resp.json()
        return body["rc"], body["stdout"], body["stderr"]

    def read_file(self, node: str, path: str) -> str:
        resp = self._http.get(self._url(node, "/agent/file"), params={"path": path})
        if resp.status_code == 404:
            raise FileNotFoundError(path)
        if resp.status_code == 403:
            raise PermissionError(path)
        resp.raise_for_status()
        return resp.text

    def write_file(self, node: str, path: str, data: str) -> None:
        resp = self._http.put(
            self._url(node, "/agent/file"), params={"path": path}, content=data
        )
        if resp.status_code == 403:
            raise PermissionError(path)
        resp.raise_for_status()

    def list_dir(self, node: str, path: str) -> List[str]:
        resp = self._http.get(self._url(node, "/agent/dir"), params={"path": path})
        if resp.status_code == 404:
            raise FileNotFoundError(path)
        if resp.status_code == 403:
            raise PermissionError(path)
        resp.raise_for_status()
        return resp.json()["entries"]

    def path_exists(self, node: str, path: str) -> bool:
        resp = self._http.get(self._url(node, "/agent/exists"), params={"path": path})
        resp.raise_for_status()
        return resp.json()["exists"]

    def close(self) -> None:
        self._http.close()
