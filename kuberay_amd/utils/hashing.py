"""Spec hashing for upgrade detection (reference: utils/util.go:620-668).

``hash_without_replicas_and_workers_to_delete`` mutes the fields the
autoscaler (Replicas / ScaleStrategy.WorkersToDelete / min / max) and
external controllers (tolerations, scheduling gates) are allowed to touch,
so that scale events never trigger a zero-downtime cluster swap.
"""
from __future__ import annotations

import base64
import hashlib
import json
from typing import Any


def json_hash(obj: Any) -> str:
    """Deterministic short hash of a JSON-serializable object
    (util.go GenerateJsonHash analog: sha1 → base32, truncated)."""
    data = json.dumps(obj, sort_keys=True, separators=(",", ":"), default=str)
    digest = hashlib.sha1(data.encode()).digest()
    return base64.b32encode(digest).decode().lower().rstrip("=")[:27]


def hash_without_replicas_and_workers_to_delete(cluster_spec) -> str:
    """util.go:645 GenerateHashWithoutReplicasAndWorkersToDelete."""
    spec = cluster_spec.clone()
    spec.upgrade_strategy = None
    spec.head_group_spec.template.spec.tolerations = None
    if hasattr(spec.head_group_spec.template.spec, "scheduling_gates"):
        spec.head_group_spec.template.spec.scheduling_gates = None  # type: ignore[attr-defined]
    for group in spec.worker_group_specs:
        group.replicas = None
        group.min_replicas = None
        group.max_replicas = None
        group.scale_strategy.workers_to_delete = None
        group.template.spec.tolerations = None
        if hasattr(group.template.spec, "scheduling_gates"):
            group.template.spec.scheduling_gates = None  # type: ignore[attr-defined]
    # extra="allow" may have captured schedulingGates as a raw field
    def _strip_gates(model):
        extra = getattr(model, "__pydantic_extra__", None)
        if extra and "schedulingGates" in extra:
            extra.pop("schedulingGates")

    _strip_gates(spec.head_group_spec.template.spec)
    for group in spec.worker_group_specs:
        _strip_gates(group.template.spec)
    return json_hash(spec.to_dict())
