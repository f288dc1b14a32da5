"""CRD YAML generation (the kubebuilder codegen analog; reference CRDs at
ray-operator/config/crd/bases/ray.io_*.yaml).

Full structural openAPIV3Schemas are generated from the pydantic models
(kuberay_amd.schema): every CR field is typed; the embedded pod-template /
core-v1 subtrees carry ``x-kubernetes-preserve-unknown-fields`` (they hold
arbitrary core/v1 content). Printer columns match the reference.
"""
from __future__ import annotations

from typing import Any, Dict, List

import yaml

from .utils import constants as C

_PRESERVE = {"type": "object", "x-kubernetes-preserve-unknown-fields": True}

_MODELS = None


def _model_for(kind: str):
    global _MODELS
    if _MODELS is None:
        from .models import RayCluster, RayCronJob, RayJob, RayService
        _MODELS = {"RayCluster": RayCluster, "RayJob": RayJob,
                   "RayService": RayService, "RayCronJob": RayCronJob}
    return _MODELS[kind]


def _crd(kind: str, plural: str, printer_columns: List[Dict[str, Any]],
         extra_names: Dict[str, Any] = None) -> Dict[str, Any]:
    from .schema import structural_schema
    schema = structural_schema(_model_for(kind))
    singular = kind.lower()
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.{C.GROUP}"},
        "spec": {
            "group": C.GROUP,
            "names": {
                "kind": kind,
                "listKind": f"{kind}List",
                "plural": plural,
                "singular": singular,
                **(extra_names or {}),
            },
            "scope": "Namespaced",
            "versions": [{
                "name": C.VERSION,
                "served": True,
                "storage": True,
                "subresources": {"status": {}},
                "additionalPrinterColumns": printer_columns,
                "schema": {"openAPIV3Schema": schema},
            }],
        },
    }


def raycluster_crd() -> Dict[str, Any]:
    # printer columns mirror raycluster_types.go kubebuilder markers
    # (minus the TPU column — MI355X scoping)
    return _crd("RayCluster", "rayclusters", [
        {"name": "desired workers", "type": "integer",
         "jsonPath": ".status.desiredWorkerReplicas"},
        {"name": "available workers", "type": "integer",
         "jsonPath": ".status.availableWorkerReplicas"},
        {"name": "cpus", "type": "string", "jsonPath": ".status.desiredCPU"},
        {"name": "memory", "type": "string", "jsonPath": ".status.desiredMemory"},
        {"name": "gpus", "type": "string", "jsonPath": ".status.desiredGPU"},
        {"name": "status", "type": "string", "jsonPath": ".status.state"},
        {"name": "age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ], extra_names={"categories": ["all"]})


def rayjob_crd() -> Dict[str, Any]:
    return _crd("RayJob", "rayjobs", [
        {"name": "job status", "type": "string", "jsonPath": ".status.jobStatus"},
        {"name": "deployment status", "type": "string",
         "jsonPath": ".status.jobDeploymentStatus"},
        {"name": "ray cluster name", "type": "string",
         "jsonPath": ".status.rayClusterName"},
        {"name": "start time", "type": "string", "jsonPath": ".status.startTime"},
        {"name": "end time", "type": "string", "jsonPath": ".status.endTime"},
        {"name": "age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ], extra_names={"categories": ["all"]})


def rayservice_crd() -> Dict[str, Any]:
    return _crd("RayService", "rayservices", [
        {"name": "service status", "type": "string",
         "jsonPath": ".status.serviceStatus"},
        {"name": "num serve endpoints", "type": "string",
         "jsonPath": ".status.numServeEndpoints"},
        {"name": "age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ], extra_names={"categories": ["all"]})


def raycronjob_crd() -> Dict[str, Any]:
    return _crd("RayCronJob", "raycronjobs", [
        {"name": "schedule", "type": "string", "jsonPath": ".spec.schedule"},
        {"name": "last schedule", "type": "string",
         "jsonPath": ".status.lastScheduleTime"},
        {"name": "age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ])


def all_crds() -> List[Dict[str, Any]]:
    return [raycluster_crd(), rayjob_crd(), rayservice_crd(), raycronjob_crd()]


def write_crds(directory: str) -> List[str]:
    import os
    os.makedirs(directory, exist_ok=True)
    out = []
    for crd in all_crds():
        plural = crd["spec"]["names"]["plural"]
        path = os.path.join(directory, f"ray.io_{plural}.yaml")
        with open(path, "w") as f:
            yaml.safe_dump(crd, f, sort_keys=False)
        out.append(path)
    return out


if __name__ == "__main__":
    import sys
    target = sys.argv[1] if len(sys.argv) > 1 else "deploy/crds"
    for p in write_crds(target):
        print(p)
