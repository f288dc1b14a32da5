"""xGMI topology discovery for gang scheduling.

On an 8xMI355X node every GPU has 7 point-to-point xGMI links (~153 GB/s
each) — an all-to-all island. Gang placement wants whole xGMI islands, and
multi-node setups want worker groups pinned inside one island. Discovery
runs ``rocm-smi --showtopo``; parsing is pure and unit-testable.
"""
from __future__ import annotations

import re
import subprocess
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

from ..utils import constants as C

XGMI_LINK_GB_S = 153.0  # per link, per direction


@dataclass
class XgmiTopology:
    num_gpus: int
    # link_type[(i, j)] = "XGMI" | "PCIE" | ...
    link_type: Dict[Tuple[int, int], str] = field(default_factory=dict)
    hops: Dict[Tuple[int, int], int] = field(default_factory=dict)

    def is_xgmi(self, a: int, b: int) -> bool:
        return self.link_type.get((a, b), "").upper().startswith("XGMI")

    def islands(self) -> List[Set[int]]:
        """Connected components over xGMI links."""
        seen: Set[int] = set()
        out: List[Set[int]] = []
        for start in range(self.num_gpus):
            if start in seen:
                continue
            comp = {start}
            frontier = [start]
            while frontier:
                cur = frontier.pop()
                for other in range(self.num_gpus):
                    if other not in comp and (self.is_xgmi(cur, other)
                                              or self.is_xgmi(other, cur)):
                        comp.add(other)
                        frontier.append(other)
            seen |= comp
            out.append(comp)
        return out

    def fully_connected(self) -> bool:
        islands = self.islands()
        return len(islands) == 1 and len(islands[0]) == self.num_gpus

    def bisection_bandwidth_gb_s(self) -> float:
        """Ring-collective per-link bound sanity number for bucket sizing."""
        xgmi_links = sum(1 for (a, b), t in self.link_type.items()
                         if a < b and t.upper().startswith("XGMI"))
        return xgmi_links * XGMI_LINK_GB_S


def parse_showtopo_text(text: str) -> XgmiTopology:
    """Parse ``rocm-smi --showtopo`` matrix sections (Link Type + Hops)."""
    link_type: Dict[Tuple[int, int], str] = {}
    hops: Dict[Tuple[int, int], int] = {}
    num_gpus = 0
    section = None
    for line in text.splitlines():
        lower = line.lower()
        if "link type between two gpus" in lower:
            section = "link"
            continue
        if "hops between two gpus" in lower:
            section = "hops"
            continue
        if lower.startswith("==") or not line.strip():
            if "====" in line and section and "gpu" not in lower:
                pass
            continue
        m = re.match(r"\s*GPU(\d+)\s+(.*)", line)
        if not m or section is None:
            continue
        row = int(m.group(1))
        cells = m.group(2).split()
        num_gpus = max(num_gpus, row + 1, len(cells))
        for col, cell in enumerate(cells):
            if row == col:
                continue
            if section == "link":
                link_type[(row, col)] = cell
            elif section == "hops":
                try:
                    hops[(row, col)] = int(cell)
                except ValueError:
                    pass
    return XgmiTopology(num_gpus=num_gpus, link_type=link_type, hops=hops)


def discover(timeout: float = 10.0) -> Optional[XgmiTopology]:
    """Run rocm-smi --showtopo on this node; None when unavailable."""
    import os
    bin_ = C.ROCM_SMI_BIN if os.path.exists(C.ROCM_SMI_BIN) else "rocm-smi"
    try:
        out = subprocess.run([bin_, "--showtopo"], capture_output=True,
                             text=True, timeout=timeout)
    except (OSError, subprocess.TimeoutExpired):
        return None
    if out.returncode != 0:
        return None
    return parse_showtopo_text(out.stdout)
