"""kuberay_amd — an MI355X-native Kubernetes operator framework for Ray.

A from-scratch rebuild of the capabilities of ray-project/kuberay (the
reference) designed MI355X-first:

* the control plane is a Python + native C++ reconcile engine (there is no
  Go toolchain in this image; the reference's Go/controller-runtime stack is
  replaced by an idiomatic Python controller runtime backed by a C++
  object-cache/workqueue core in ``kuberay_amd._native``),
* all GPU specificity lives in pod construction, health probing and scale
  decisions: worker pods request ``amd.com/gpu`` (AMD k8s device plugin),
  mount ``/dev/kfd`` + ``/dev/dri``, pin ROCm images, and get RCCL-over-xGMI
  environment injected (reference analog: ray-operator/controllers/ray/common/pod.go),
* readiness probing of GPU workers goes beyond HTTP checks: a gfx950 HIP
  extension (``kuberay_amd._native.gpuhealth``) launches a real MFMA smoke
  kernel and an HBM bandwidth probe on-device,
* the autoscaler decision loop reads rocm-smi utilisation / HBM occupancy
  (288 GB per GPU) instead of nvidia-smi, and gang scheduling is
  xGMI-topology aware.

There is no NVIDIA/CUDA code path anywhere — no dual-vendor dispatch.
"""

__version__ = "0.1.0"

KUBERAY_VERSION = __version__
