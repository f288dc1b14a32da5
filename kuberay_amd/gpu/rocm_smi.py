"""rocm-smi / amd-smi interrogation for MI355X nodes.

The MI355X replacement for everything nvidia-smi-shaped: GPU utilisation and
HBM occupancy (288 GB per GPU) feed the autoscaler decision loop
(kuberay_amd/gpu/autoscaler.py) and readiness probing
(kuberay_amd/gpu/health.py). Parsing is separated from process execution so
it is unit-testable on CPU-only hosts with canned JSON.
"""
from __future__ import annotations

import json
import os
import shutil
import subprocess
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..utils import constants as C

MI355X_HBM_BYTES = 288 * 1024**3


@dataclass
class GpuStats:
    index: int
    utilization_pct: float = 0.0        # GPU use (%)
    vram_used_bytes: int = 0
    vram_total_bytes: int = MI355X_HBM_BYTES
    temperature_c: Optional[float] = None
    power_w: Optional[float] = None

    @property
    def vram_used_fraction(self) -> float:
        if self.vram_total_bytes <= 0:
            return 0.0
        return self.vram_used_bytes / self.vram_total_bytes


def rocm_smi_available() -> bool:
    return os.path.exists(C.ROCM_SMI_BIN) or shutil.which("rocm-smi") is not None


def _rocm_smi_bin() -> str:
    return C.ROCM_SMI_BIN if os.path.exists(C.ROCM_SMI_BIN) else "rocm-smi"


def query_rocm_smi_json(timeout: float = 10.0) -> Dict:
    """Run rocm-smi for use%, VRAM, temperature and power in one JSON call."""
    cmd = [_rocm_smi_bin(), "--showuse", "--showmeminfo", "vram",
           "--showtemp", "--showpower", "--json"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
    if out.returncode != 0:
        raise RuntimeError(f"rocm-smi failed rc={out.returncode}: {out.stderr[:400]}")
    # rocm-smi sometimes prefixes warnings (e.g. low-power state) — take the
    # JSON from the first brace
    text = out.stdout
    start = text.find("{")
    if start < 0:
        raise RuntimeError(f"rocm-smi produced no JSON: {text[:200]}")
    return json.loads(text[start:])


def parse_rocm_smi_json(data: Dict) -> List[GpuStats]:
    """Parse rocm-smi --json output (keys like "card0", "card1", ...)."""
    stats: List[GpuStats] = []
    for key in sorted(data):
        if not key.startswith("card"):
            continue
        card = data[key]
        try:
            index = int(key[4:])
        except ValueError:
            continue
        s = GpuStats(index=index)
        for k, v in card.items():
            lk = k.lower()
            try:
                if "gpu use" in lk:
                    # NOTE: "GFX Activity" is a monotonically increasing
                    # counter, NOT a percentage — never treat it as one.
                    s.utilization_pct = float(v)
                elif "vram" in lk and "used" in lk:
                    s.vram_used_bytes = int(v)
                elif "vram" in lk and "total" in lk:
                    s.vram_total_bytes = int(v)
                elif "temperature" in lk and ("junction" in lk or "edge" in lk):
                    s.temperature_c = float(v)
                elif "average graphics package power" in lk or \
                        "current socket graphics package power" in lk:
                    s.power_w = float(v)
            except (TypeError, ValueError):
                continue
        stats.append(s)
    return stats


def get_gpu_stats(timeout: float = 10.0) -> List[GpuStats]:
    return parse_rocm_smi_json(query_rocm_smi_json(timeout))


def node_gpu_summary(stats: Optional[List[GpuStats]] = None) -> Dict:
    """Aggregate stats the autoscaler consumes."""
    if stats is None:
        stats = get_gpu_stats()
    if not stats:
        return {"gpu_count": 0, "avg_utilization_pct": 0.0,
                "max_utilization_pct": 0.0, "avg_vram_used_fraction": 0.0,
                "max_vram_used_fraction": 0.0}
    return {
        "gpu_count": len(stats),
        "avg_utilization_pct": sum(s.utilization_pct for s in stats) / len(stats),
        "max_utilization_pct": max(s.utilization_pct for s in stats),
        "avg_vram_used_fraction": sum(s.vram_used_fraction for s in stats) / len(stats),
        "max_vram_used_fraction": max(s.vram_used_fraction for s in stats),
    }
