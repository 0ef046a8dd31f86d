"""HTTP admission server for real-cluster deployments.

Serves the Kubernetes ValidatingWebhookConfiguration endpoint
``/validate-cro-hpsys-ibm-ie-com-v1alpha1-composabilityrequest`` (path parity
with the kubebuilder marker, composabilityrequest_webhook.go:49) speaking the
``admission.k8s.io/v1`` AdmissionReview protocol, plus ``/healthz`` and
``/readyz`` (cmd/main.go:205-212 parity).

The rule logic is shared with the in-process validator
(cro_amd/webhook/validator.py); this module only adapts it to
AdmissionReview request/response framing.  ``existing_requests_fn`` supplies
the current ComposabilityRequests (a runtime Client lookup in-process; an
apiserver LIST in cluster mode).
"""

from __future__ import annotations

from typing import Callable, List

from fastapi import FastAPI, Request

from ..api.v1alpha1.types import ComposabilityRequest
from .validator import validate_composability_request

WEBHOOK_PATH = "/validate-cro-hpsys-ibm-ie-com-v1alpha1-composabilityrequest"


def build_app(existing_requests_fn: Callable[[], List[ComposabilityRequest]]) -> FastAPI:
    app = FastAPI(title="cro-amd admission webhook")

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get("/readyz")
    def readyz():
        return {"status": "ok"}

    @app.post(WEBHOOK_PATH)
    async def validate(request: Request):
        review = await request.json()
        req = review.get("request", {})
        uid = req.get("uid", "")
        operation = req.get("operation", "")

        allowed, message = True, ""
        if operation in ("CREATE", "UPDATE"):
            try:
                obj = ComposabilityRequest.model_validate(req.get("object", {}))
            except Exception as exc:
                allowed, message = False, f"invalid ComposabilityRequest: {exc}"
            else:
                msg = validate_composability_request(obj, existing_requests_fn())
                if msg:
                    allowed, message = False, msg

        response = {"uid": uid, "allowed": allowed}
        if not allowed:
            response["status"] = {"message": message, "code": 403}
        return {
            "apiVersion": "admission.k8s.io/v1",
            "kind": "AdmissionReview",
            "response": response,
        }

    return app


def serve(existing_requests_fn, host: str = "0.0.0.0", port: int = 9443, **uvicorn_kwargs):
    """Run the admission server (TLS material via uvicorn kwargs
    ``ssl_certfile``/``ssl_keyfile`` — cert-manager mounts them in cluster)."""
    import uvicorn

    uvicorn.run(build_app(existing_requests_fn), host=host, port=port, **uvicorn_kwargs)
