"""MI355X GPU layer: rocm-smi telemetry, on-device health probes (gfx950 HIP
extension), xGMI topology discovery, and the rocm-smi-driven autoscaler."""
