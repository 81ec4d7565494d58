"""Admission webhook server — parity with the reference's knative-style
validation webhooks (pkg/workspace/webhooks/webhooks.go:41-59 registers
controllers on /validate/workspace.kaito.sh etc.; deep semantics in
api/*/..._validation.go). Serves AdmissionReview v1: validates the typed
CRD objects and returns allowed/denied with the validation message.
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

from fastapi import FastAPI, Request

from .api_types import ValidationError
from .main import workspace_from_obj
from .sku import CloudSKUHandler


def _deny(uid: str, msg: str) -> Dict:
    return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
            "response": {"uid": uid, "allowed": False,
                         "status": {"message": msg, "code": 400}}}


def _allow(uid: str) -> Dict:
    return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
            "response": {"uid": uid, "allowed": True}}


def build_webhook_app(sku_handler: Optional[CloudSKUHandler] = None,
                      known_presets=None) -> FastAPI:
    app = FastAPI(title="kaito-amd admission webhooks")

    def _validate_workspace(obj: Dict) -> None:
        ws = workspace_from_obj(obj)
        ws.validate(sku_handler=sku_handler, known_presets=known_presets)

    def _validate_inferenceset(obj: Dict) -> None:
        spec = obj.get("spec", {})
        tpl = spec.get("workspaceTemplate")
        if tpl is None:
            raise ValidationError("workspaceTemplate required")
        if spec.get("replicas", 1) < 0:
            raise ValidationError("replicas must be >= 0")
        _validate_workspace({"metadata": obj.get("metadata", {}),
                             "spec": tpl})

    validators: Dict[str, Callable[[Dict], None]] = {
        "workspace.kaito.sh": _validate_workspace,
        "inferenceset.kaito.sh": _validate_inferenceset,
    }

    @app.post("/validate/{group}")
    async def validate(group: str, request: Request):
        body = await request.json()
        req = body.get("request", {})
        uid = req.get("uid", "")
        fn = validators.get(group)
        if fn is None:
            return _deny(uid, f"no validator for {group}")
        try:
            fn(req.get("object", {}))
        except (ValidationError, KeyError, TypeError) as e:
            return _deny(uid, str(e))
        return _allow(uid)

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    return app
