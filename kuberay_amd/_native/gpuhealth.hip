// gpuhealth — on-device MI355X (gfx950) health probe kernels + pybind11 host.
//
// Used by the operator's GPU worker readiness/liveness gate
// (kuberay_amd/gpu/health.py, wired into pod readiness probes by
// kuberay_amd/common/pod.py). The reference operator (kuberay) has no
// GPU-level probing at all; on MI355X nodes a pod can be "Ready" while its
// GPU is wedged (queue hang, HBM fallout, xGMI link down). These probes are
// cheap (<50 ms) and catch that:
//
//   * mfma_smoke   — one wave per workgroup issues v_mfma_f32_16x16x32_bf16
//                    with all-ones inputs; every D element must equal K=32.
//                    Proves the matrix pipeline executes and returns.
//   * hbm_stream   — float4 streaming copy sized ≫ L3 (256 MiB) across
//                    ≫256 workgroups; host computes GB/s and compares
//                    against a floor (healthy MI355X streams ≈6 TB/s; a
//                    sick HBM stack or thermally-parked clock shows up as a
//                    collapse by an order of magnitude).
//
// Build: hipcc --offload-arch=gfx950 (see kuberay_amd/_native/build.py).
// Wavefront width is 64 on CDNA4 — all wave math below hard-codes 64.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " #expr);          \
    }                                                                          \
  } while (0)

// ---------------------------------------------------------------------------
// MFMA smoke: D = A(1s) * B(1s) + 0  ->  every element == K == 32
// ---------------------------------------------------------------------------
#if defined(__HIP_DEVICE_COMPILE__) && !defined(__gfx950__)
// Only gfx950 is supported — refuse silently-wrong builds for other arches.
#error "gpuhealth targets gfx950 (MI355X) only"
#endif

typedef __attribute__((ext_vector_type(8))) __bf16 frag_ab_t;  // 8 bf16 = 4 VGPRs
typedef __attribute__((ext_vector_type(4))) float frag_cd_t;   // 4 fp32 acc

__global__ void mfma_smoke_kernel(float* __restrict__ out) {
  frag_ab_t a, b;
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)1.0f;
    b[i] = (__bf16)1.0f;
  }
  frag_cd_t acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  const int lane = threadIdx.x & 63;
  const int wg = blockIdx.x;
  // 4 accumulator elements per lane; layout-independent check (all == 32)
  for (int r = 0; r < 4; ++r) {
    out[(size_t)wg * 256 + lane * 4 + r] = acc[r];
  }
}

static bool run_mfma_smoke(int workgroups) {
  const size_t n = (size_t)workgroups * 256;
  float* d_out = nullptr;
  HIP_CHECK(hipMalloc(&d_out, n * sizeof(float)));
  hipLaunchKernelGGL(mfma_smoke_kernel, dim3(workgroups), dim3(64), 0, 0, d_out);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  std::vector<float> host(n);
  HIP_CHECK(hipMemcpy(host.data(), d_out, n * sizeof(float), hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(d_out));
  for (size_t i = 0; i < n; ++i) {
    if (host[i] != 32.0f) return false;
  }
  return true;
}

// ---------------------------------------------------------------------------
// HBM streaming copy (float4, grid-stride) — measures achievable bandwidth
// ---------------------------------------------------------------------------
__global__ void hbm_stream_kernel(const float4* __restrict__ src,
                                  float4* __restrict__ dst, size_t n_vec) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n_vec; i += stride) {
    dst[i] = src[i];
  }
}

static double run_hbm_stream(size_t bytes, int iters) {
  const size_t n_vec = bytes / sizeof(float4);
  float4 *d_src = nullptr, *d_dst = nullptr;
  HIP_CHECK(hipMalloc(&d_src, n_vec * sizeof(float4)));
  HIP_CHECK(hipMalloc(&d_dst, n_vec * sizeof(float4)));
  HIP_CHECK(hipMemset(d_src, 1, n_vec * sizeof(float4)));

  // ≫256 workgroups to fill 8 XCDs (guide §1); 256 threads/block.
  const int blocks = 4096;
  const int threads = 256;

  // warmup
  hipLaunchKernelGGL(hbm_stream_kernel, dim3(blocks), dim3(threads), 0, 0,
                     d_src, d_dst, n_vec);
  HIP_CHECK(hipDeviceSynchronize());

  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0, 0));
  for (int it = 0; it < iters; ++it) {
    hipLaunchKernelGGL(hbm_stream_kernel, dim3(blocks), dim3(threads), 0, 0,
                       d_src, d_dst, n_vec);
  }
  HIP_CHECK(hipEventRecord(t1, 0));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  HIP_CHECK(hipFree(d_src));
  HIP_CHECK(hipFree(d_dst));
  // read + write traffic
  const double total_bytes = 2.0 * (double)n_vec * sizeof(float4) * iters;
  return total_bytes / (ms * 1e-3) / 1e9;  // GB/s
}

// ---------------------------------------------------------------------------
// host API
// ---------------------------------------------------------------------------
static int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

static py::dict device_info(int device) {
  HIP_CHECK(hipSetDevice(device));
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  py::dict d;
  d["name"] = std::string(prop.name);
  d["gcn_arch"] = std::string(prop.gcnArchName);
  d["total_mem_gb"] = (double)prop.totalGlobalMem / (1024.0 * 1024.0 * 1024.0);
  d["multi_processor_count"] = prop.multiProcessorCount;
  d["warp_size"] = prop.warpSize;
  d["clock_mhz"] = prop.clockRate / 1000;
  return d;
}

static bool mfma_smoke(int device, int workgroups) {
  HIP_CHECK(hipSetDevice(device));
  return run_mfma_smoke(workgroups);
}

static double hbm_bandwidth_gb_s(int device, double gib, int iters) {
  HIP_CHECK(hipSetDevice(device));
  size_t bytes = (size_t)(gib * 1024.0 * 1024.0 * 1024.0);
  if (bytes < (64u << 20)) bytes = (64u << 20);
  return run_hbm_stream(bytes, iters);
}

static py::dict health_check(int device, double hbm_floor_gb_s, bool quick) {
  py::dict result;
  HIP_CHECK(hipSetDevice(device));
  bool mfma_ok = run_mfma_smoke(quick ? 256 : 2048);
  // quick: 0.5 GiB per buffer (still > L2, probes real HBM); full: 4 GiB
  // (> 256 MiB L3 several times over per pass).
  double bw = run_hbm_stream(quick ? (512ull << 20) : (4ull << 30), quick ? 3 : 10);
  result["mfma_ok"] = mfma_ok;
  result["hbm_gb_s"] = bw;
  result["hbm_ok"] = bw >= hbm_floor_gb_s;
  result["healthy"] = mfma_ok && bw >= hbm_floor_gb_s;
  return result;
}

PYBIND11_MODULE(gpuhealth, m) {
  m.doc() = "MI355X (gfx950) on-device health probes: MFMA smoke + HBM stream";
  m.def("device_count", &device_count);
  m.def("device_info", &device_info, py::arg("device") = 0);
  m.def("mfma_smoke", &mfma_smoke, py::arg("device") = 0,
        py::arg("workgroups") = 2048);
  m.def("hbm_bandwidth_gb_s", &hbm_bandwidth_gb_s, py::arg("device") = 0,
        py::arg("gib") = 1.0, py::arg("iters") = 10);
  m.def("health_check", &health_check, py::arg("device") = 0,
        py::arg("hbm_floor_gb_s") = 1000.0, py::arg("quick") = false);
}
