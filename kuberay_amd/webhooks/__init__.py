"""Validating admission webhooks (reference: ray-operator/pkg/webhooks/v1,
opt-in via ENABLE_WEBHOOKS; main.go:344-351).

``handle_admission_review`` implements the AdmissionReview v1 contract for
the three CRDs, backed by the same pure validation functions the
reconcilers use. ``create_webhook_app`` serves it over FastAPI at
``/validate-ray-io-v1-{kind}``.
"""
from __future__ import annotations

from typing import Any, Dict

from ..models import RayCluster, RayJob, RayService
from ..utils.validation import (
    validate_raycluster_metadata,
    validate_raycluster_spec,
    validate_rayjob_metadata,
    validate_rayjob_spec,
    validate_rayservice_metadata,
    validate_rayservice_spec,
)

_VALIDATORS = {
    "RayCluster": (RayCluster, lambda o: validate_raycluster_metadata(o.metadata)
                   + validate_raycluster_spec(o)),
    "RayJob": (RayJob, lambda o: validate_rayjob_metadata(o.metadata)
               + validate_rayjob_spec(o)),
    "RayService": (RayService, lambda o: validate_rayservice_metadata(o.metadata)
                   + validate_rayservice_spec(o)),
}


def handle_admission_review(review: Dict[str, Any]) -> Dict[str, Any]:
    """AdmissionReview v1 in → AdmissionReview v1 out."""
    request = review.get("request") or {}
    uid = request.get("uid", "")
    obj = request.get("object") or {}
    kind = obj.get("kind") or request.get("kind", {}).get("kind", "")

    allowed, message = True, ""
    entry = _VALIDATORS.get(kind)
    if entry is not None:
        model, validator = entry
        try:
            typed = model.from_dict(obj)
            errs = validator(typed)
        except Exception as e:  # malformed object
            errs = [f"invalid {kind}: {e}"]
        if errs:
            allowed, message = False, "; ".join(errs)

    response: Dict[str, Any] = {"uid": uid, "allowed": allowed}
    if not allowed:
        response["status"] = {"code": 400, "message": message}
    return {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
            "response": response}


def create_webhook_app():
    from fastapi import FastAPI

    app = FastAPI(title="kuberay-amd-webhooks")

    @app.post("/validate-ray-io-v1-raycluster")
    @app.post("/validate-ray-io-v1-rayjob")
    @app.post("/validate-ray-io-v1-rayservice")
    def validate(review: Dict[str, Any]):
        return handle_admission_review(review)

    return app
