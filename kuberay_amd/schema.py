"""Structural openAPIV3Schema generation from the pydantic CRD models
(the controller-gen analog; reference CRDs ship full structural schemas in
ray-operator/config/crd/bases/).

Pydantic emits JSON Schema with $defs/$ref and keywords Kubernetes rejects;
this module inlines refs, strips unsupported keywords, and marks the
embedded pod-template/service subtrees ``x-kubernetes-preserve-unknown-fields``
(they embed arbitrary core/v1 content, same approach as many production
CRDs for PodTemplateSpec).
"""
from __future__ import annotations

import copy
from typing import Any, Dict

# property names whose subtree is arbitrary core/v1 content
_PRESERVE_SUBTREES = {
    "template", "headService", "serveService", "submitterPodTemplate",
    "volumes", "affinity", "tolerations", "env", "envFrom", "volumeMounts",
    "resources", "securityContext", "ingressOptions", "valueFrom",
    "ingressRules", "egressRules", "imagePullSecrets", "lifecycle",
    "livenessProbe", "readinessProbe", "startupProbe",
}

# JSON Schema keywords the K8s structural-schema validator rejects
_STRIP_KEYWORDS = {
    "title", "default", "anyOf", "allOf", "oneOf", "const",
    "additionalProperties", "discriminator", "definitions", "$defs",
    "exclusiveMaximum", "exclusiveMinimum", "patternProperties",
}

_PRESERVE = {"type": "object", "x-kubernetes-preserve-unknown-fields": True}


def _resolve_ref(ref: str, defs: Dict[str, Any]) -> Dict[str, Any]:
    name = ref.rsplit("/", 1)[-1]
    return defs.get(name, {})


def _sanitize(node: Any, defs: Dict[str, Any], depth: int = 0,
              seen: tuple = ()) -> Any:
    if depth > 30:
        return dict(_PRESERVE)
    if isinstance(node, list):
        return [_sanitize(v, defs, depth + 1, seen) for v in node]
    if not isinstance(node, dict):
        return node

    node = dict(node)

    # inline $ref (cycle-guarded: recursive models collapse to preserve)
    if "$ref" in node:
        ref = node["$ref"]
        if ref in seen:
            return dict(_PRESERVE)
        resolved = _resolve_ref(ref, defs)
        merged = {**resolved, **{k: v for k, v in node.items() if k != "$ref"}}
        return _sanitize(merged, defs, depth + 1, seen + (ref,))

    # pydantic Optional[...] → anyOf [X, null]: take the non-null branch
    if "anyOf" in node:
        branches = [b for b in node["anyOf"]
                    if not (isinstance(b, dict) and b.get("type") == "null")]
        if len(branches) == 1:
            merged = {**branches[0],
                      **{k: v for k, v in node.items() if k != "anyOf"}}
            return _sanitize(merged, defs, depth + 1, seen)
        return dict(_PRESERVE)

    for kw in _STRIP_KEYWORDS:
        node.pop(kw, None)

    if "properties" in node:
        props = {}
        for name, sub in node["properties"].items():
            if name in _PRESERVE_SUBTREES:
                props[name] = dict(_PRESERVE)
            else:
                props[name] = _sanitize(sub, defs, depth + 1, seen)
        node["properties"] = props
        node.setdefault("type", "object")
    if "items" in node:
        node["items"] = _sanitize(node["items"], defs, depth + 1, seen)
    # integer/number bounds survive; enum survives; pattern survives
    return node


def structural_schema(model) -> Dict[str, Any]:
    """Full openAPIV3Schema for a CRD model (spec+status typed)."""
    raw = model.model_json_schema(ref_template="#/$defs/{model}")
    defs = raw.get("$defs", {})
    out = _sanitize(raw, defs)
    # top level: apiVersion/kind/metadata/spec/status
    props = out.get("properties", {})
    props["metadata"] = {"type": "object"}
    props.pop("apiVersion", None)
    props.pop("kind", None)
    props["apiVersion"] = {"type": "string"}
    props["kind"] = {"type": "string"}
    return {"type": "object", "properties": props}
