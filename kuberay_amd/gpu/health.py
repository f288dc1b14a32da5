"""GPU health gate: rocm-smi liveness + on-device gfx950 probes.

Backs the worker readiness probe injected by the pod builder
(``python -m kuberay_amd.gpu.probe``) and the GPU-gated sim-kubelet used in
``pytest -m gpu`` runs. The native extension is REQUIRED on GPU hosts — if
/dev/kfd exists and the extension is missing, this module raises instead of
silently passing (no silent eager fallback).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

from . import rocm_smi

# Healthy MI355X streams ~6 TB/s; 1 TB/s floor catches an HBM stack or
# clock-park collapse without flaking on a busy GPU.
DEFAULT_HBM_FLOOR_GB_S = 1000.0


class GpuHealthError(RuntimeError):
    pass


def gpu_node() -> bool:
    """Is this host a ROCm GPU node?"""
    return os.path.exists("/dev/kfd")


def _load_native():
    try:
        from .._native import gpuhealth  # type: ignore
        return gpuhealth
    except ImportError as e:
        if gpu_node():
            raise GpuHealthError(
                "kuberay_amd._native.gpuhealth extension is not built on a GPU "
                "node — build it with `python -m kuberay_amd._native.build` "
                f"(import error: {e})") from e
        return None


@dataclass
class HealthReport:
    healthy: bool
    mfma_ok: Optional[bool] = None
    hbm_gb_s: Optional[float] = None
    rocm_smi_ok: Optional[bool] = None
    detail: str = ""


def check_gpu_health(device: int = 0, quick: bool = False,
                     hbm_floor_gb_s: float = DEFAULT_HBM_FLOOR_GB_S) -> HealthReport:
    """Full gate: rocm-smi responds AND the device executes MFMA AND streams
    HBM above the floor. Raises GpuHealthError on non-GPU hosts."""
    if not gpu_node():
        raise GpuHealthError("not a GPU node (/dev/kfd missing)")

    rocm_ok = True
    try:
        stats = rocm_smi.get_gpu_stats()
        if not stats:
            rocm_ok = False
    except Exception:
        rocm_ok = False

    native = _load_native()
    if native.device_count() <= device:
        return HealthReport(healthy=False, rocm_smi_ok=rocm_ok,
                            detail=f"device {device} not visible to HIP")
    result = native.health_check(device, hbm_floor_gb_s, quick)
    return HealthReport(
        healthy=bool(result["healthy"]) and rocm_ok,
        mfma_ok=bool(result["mfma_ok"]),
        hbm_gb_s=float(result["hbm_gb_s"]),
        rocm_smi_ok=rocm_ok,
        detail="" if result["healthy"] else
        f"mfma_ok={result['mfma_ok']} hbm_gb_s={result['hbm_gb_s']:.0f}",
    )


def sim_kubelet_gpu_gate(pod: dict) -> bool:
    """gpu_gate hook for SimKubelet: a GPU-requesting pod only turns Ready if
    the local MI355X passes the quick health gate."""
    try:
        return check_gpu_health(quick=True).healthy
    except GpuHealthError:
        return False
