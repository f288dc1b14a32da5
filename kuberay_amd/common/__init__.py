"""Pure-function builders: pods, services, submitter jobs, RBAC, ingress,
GCS-FT cleanup. All MI355X specificity (amd.com/gpu, RCCL env, /dev/kfd,
rocm-smi probes) lands here."""
