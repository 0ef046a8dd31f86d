"""Node-agent HTTP API: remote node access for off-node controllers.

The reference reaches node hardware by SPDY-exec'ing shell pipelines into
privileged pods (gpus.go:1040-1067).  Here the node agent is a first-class
process; controllers running off-node reach its NodeExec surface over this
API (cro_amd/runtime exposes the matching client, RemoteNodeExec):

    POST /agent/run        {"argv": [...], "timeout": s} → {rc, stdout, stderr}
    GET  /agent/file?path= → raw contents          (404 on missing)
    PUT  /agent/file?path= (body = contents)
    GET  /agent/dir?path=  → {"entries": [...]}
    GET  /agent/exists?path= → {"exists": bool}

Only the NodeExec verbs are exposed — no arbitrary shell.  Binary execution
inherits LocalNodeExec's trusted-path resolution (never $PATH).

Authentication: set ``CRO_AGENT_TOKEN`` (or pass ``token=``) and every
/agent route requires ``Authorization: Bearer <token>`` — the standalone
analog of the RBAC that gated the reference's pods/exec path
(config/rbac/role.yaml pods/exec verbs). RemoteNodeExec sends the token
from the same env var.

FAIL-CLOSED: building the agent surface with no token configured raises
``AgentAuthError`` — these routes execute binaries and write files as the
operator user, so an unauthenticated surface bound to 0.0.0.0 would be
remote code execution. Tests and loopback-only deployments may opt out
explicitly with ``allow_insecure=True``; the production entrypoint
auto-generates a token instead (cmd/main.py).
"""

from __future__ import annotations

import hmac
import os

from fastapi import FastAPI, HTTPException, Request, Response

from ..nodeops.execs import ExecError, NodeExec


class AgentAuthError(RuntimeError):
    """No agent token configured and insecure mode not explicitly allowed."""


def build_agent_app(
    execer: NodeExec, node_name: str = "local", app: FastAPI = None,
    token: str = None, allow_insecure: bool = False,
) -> FastAPI:
    """Build the agent app, or graft the /agent routes onto an existing app
    (the operator entrypoint serves API + agent surface in one process).

    Refuses to register the routes when no bearer token is configured
    (fail closed — see module docstring) unless ``allow_insecure=True``.
    """
    if token is None:
        token = os.environ.get("CRO_AGENT_TOKEN", "")
    if not token and not allow_insecure:
        raise AgentAuthError(
            "refusing to serve the /agent API without a bearer token: set "
            "CRO_AGENT_TOKEN (or pass token=), or pass allow_insecure=True "
            "for loopback-only test deployments"
        )

    def authorize(request: Request) -> None:
        if not token:
            return
        auth = request.headers.get("authorization", "")
        if not hmac.compare_digest(auth, f"Bearer {token}"):
            raise HTTPException(401, "agent API requires a valid bearer token")

    standalone = app is None
    if standalone:
        app = FastAPI(title="cro-amd node agent")

        @app.get("/healthz")
        def healthz():
            return {"status": "ok", "node": node_name}

    @app.post("/agent/run")
    async def run(request: Request):
        authorize(request)
        body = await request.json()
        argv = body.get("argv", [])
        if not argv or not isinstance(argv, list):
            raise HTTPException(422, "argv must be a non-empty list")
        timeout = float(body.get("timeout", 60.0))
        try:
            rc, out, err = execer.run(node_name, [str(a) for a in argv], timeout=timeout)
        except ExecError as exc:
            raise HTTPException(400, str(exc))
        return {"rc": rc, "stdout": out, "stderr": err}

    @app.get("/agent/file")
    def read_file(path: str, request: Request):
        authorize(request)
        try:
            return Response(execer.read_file(node_name, path), media_type="text/plain")
        except FileNotFoundError:
            raise HTTPException(404, f"{path} not found")
        except (PermissionError, OSError) as exc:
            raise HTTPException(403, str(exc))

    @app.put("/agent/file")
    async def write_file(path: str, request: Request):
        authorize(request)
        data = (await request.body()).decode()
        try:
            execer.write_file(node_name, path, data)
        except (PermissionError, OSError) as exc:
            raise HTTPException(403, str(exc))
        return {"written": path}

    @app.get("/agent/dir")
    def list_dir(path: str, request: Request):
        authorize(request)
        try:
            return {"entries": execer.list_dir(node_name, path)}
        except FileNotFoundError:
            raise HTTPException(404, f"{path} not found")
        except (PermissionError, OSError) as exc:
            raise HTTPException(403, str(exc))

    @app.get("/agent/exists")
    def exists(path: str, request: Request):
        authorize(request)
        return {"exists": execer.path_exists(node_name, path)}

    return app
