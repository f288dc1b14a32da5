"""Feature gates (reference: ray-operator/pkg/features/features.go:20-117).

Same gate names and default stages as the reference snapshot; toggled via
``--feature-gates=Name=true,Other=false`` or programmatically.
"""
from __future__ import annotations

import threading
from typing import Dict

# gate -> default enabled (reference defaults :105-117)
_DEFAULTS: Dict[str, bool] = {
    "RayClusterStatusConditions": True,       # beta, on
    "RayJobDeletionPolicy": True,             # beta, on
    "RayMultiHostIndexing": True,             # beta, on
    "RayServiceIncrementalUpgrade": True,     # beta, on (features.go:105-117)
    "RayCronJob": False,              # alpha, off (features.go:110)
    "SidecarSubmitterRestart": False,
    "RayClusterNetworkPolicy": False,
    "GCSFaultToleranceEmbeddedStorage": False,  # alpha, off (features.go:113)
    "RayClusterMTLS": False,
    "RayClusterHistoryServer": False,
    "KubernetesWAS": False,
    # MI355X-native gates
    "MI355XGpuHealthProbes": True,
    "MI355XAutoscaler": True,
    "XgmiGangScheduling": True,
}

_lock = threading.Lock()
_overrides: Dict[str, bool] = {}


def enabled(gate: str) -> bool:
    with _lock:
        if gate in _overrides:
            return _overrides[gate]
    if gate not in _DEFAULTS:
        raise KeyError(f"unknown feature gate '{gate}'")
    return _DEFAULTS[gate]


def set_gate(gate: str, value: bool) -> None:
    if gate not in _DEFAULTS:
        raise KeyError(f"unknown feature gate '{gate}'")
    with _lock:
        _overrides[gate] = value


def parse_feature_gates(spec: str) -> None:
    """--feature-gates=A=true,B=false."""
    if not spec:
        return
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        if "=" not in part:
            raise ValueError(f"invalid feature gate '{part}' (want Name=bool)")
        name, _, val = part.partition("=")
        set_gate(name.strip(), val.strip().lower() == "true")


def reset() -> None:
    with _lock:
        _overrides.clear()


def all_gates() -> Dict[str, bool]:
    return {g: enabled(g) for g in _DEFAULTS}
