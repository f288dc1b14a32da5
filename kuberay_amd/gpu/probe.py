"""Readiness-probe CLI: ``python -m kuberay_amd.gpu.probe [--quick] [--device N]``.

Injected into GPU worker readiness probes by the pod builder
(common/pod.py init_liveness_and_readiness_probe). Exit 0 = healthy.
"""
from __future__ import annotations

import argparse
import json
import sys

from .health import GpuHealthError, check_gpu_health


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="MI355X GPU health probe")
    parser.add_argument("--quick", action="store_true",
                        help="fast gate for readiness probes (<~100 ms)")
    parser.add_argument("--device", type=int, default=0)
    parser.add_argument("--hbm-floor-gb-s", type=float, default=1000.0)
    parser.add_argument("--json", action="store_true", dest="json_out")
    args = parser.parse_args(argv)

    try:
        report = check_gpu_health(device=args.device, quick=args.quick,
                                  hbm_floor_gb_s=args.hbm_floor_gb_s)
    except GpuHealthError as e:
        print(f"gpu probe error: {e}", file=sys.stderr)
        return 2
    if args.json_out:
        print(json.dumps({
            "healthy": report.healthy, "mfma_ok": report.mfma_ok,
            "hbm_gb_s": report.hbm_gb_s, "rocm_smi_ok": report.rocm_smi_ok,
            "detail": report.detail}))
    else:
        print(f"healthy={report.healthy} mfma_ok={report.mfma_ok} "
              f"hbm_gb_s={report.hbm_gb_s} rocm_smi_ok={report.rocm_smi_ok}")
    return 0 if report.healthy else 1


if __name__ == "__main__":
    sys.exit(main())
