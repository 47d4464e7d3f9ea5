// gpud_amd active-diagnostic kernels for MI355X (gfx950, CDNA4).
//
// The DCGM-diag equivalent of the reference's "active diagnostic" surface
// (the reference only collects nvidia-bug-report.sh — pkg/session/
// diagnostic.go:48; per BASELINE.json the AMD build adds real device
// diagnostics). Three stress kernels, each returning a measured rate the
// health components threshold against per-board floors:
//
//   * mfma_stress_{bf16,fp8}: register-resident MFMA loop — back-to-back
//     v_mfma_f32_32x32x16_{bf16,fp8} on 4 independent accumulators per
//     wave (the 32-cycle/SIMD issue rate needs >=2 independent
//     accumulators; see MI355X_MICROARCH.md per-instruction constants).
//     All-ones inputs make the result exact: each MFMA adds K=16 to every
//     accumulator element, so after N iterations acc == 16*N — a built-in
//     numerics check (any CU computing wrong bits fails it loudly).
//     Dense bf16 ceiling ~2.5 PF; non-scaled fp8 runs at the bf16 rate
//     (cdna_hip_programming.md §3 µbench table).
//
//   * hbm_triad / hbm_read: float4 streaming (c = a + s*b and a pure
//     read-reduce), grid >> 256 workgroups to fill 8 XCDs. Achievable
//     HBM3E bandwidth ~6.3 TB/s of the 8.0 TB/s peak.
//
//   * lds_bandwidth: ds_read_b128 sweep over a 32 KiB LDS image from 8
//     waves/CU (the §LDS table needs >=4 waves issuing wide reads to
//     reach 256 B/clk/CU).
//
// Built for gfx950 only — no multi-arch dispatch (csrc/build.sh).

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string(#expr) + " failed: " +           \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

// ---------------------------------------------------------------------------
// MFMA stress
// ---------------------------------------------------------------------------

constexpr int kAccums = 4;  // independent accumulators per wave
constexpr int kInnerUnroll = 8;

__global__ __launch_bounds__(256, 4) void mfma_stress_bf16_kernel(
    float* __restrict__ out, int iters) {
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)1.0f;
    b[i] = (__bf16)1.0f;
  }
  f32x16 acc[kAccums] = {};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int u = 0; u < kInnerUnroll; ++u) {
#pragma unroll
      for (int j = 0; j < kAccums; ++j) {
        acc[j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc[j], 0, 0, 0);
      }
    }
  }
  // one representative lane-element per accumulator; every element should
  // equal 16.0f * iters * kInnerUnroll
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < kAccums; ++j) s += acc[j][0];
  out[blockIdx.x * blockDim.x + threadIdx.x] = s;
}

__global__ __launch_bounds__(256, 4) void mfma_stress_fp8_kernel(
    float* __restrict__ out, int iters) {
  // fp8 e4m3 1.0 == 0x38; pack 8 per 64-bit operand
  const long one8 = 0x3838383838383838L;
  f32x16 acc[kAccums] = {};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int u = 0; u < kInnerUnroll; ++u) {
#pragma unroll
      for (int j = 0; j < kAccums; ++j) {
        acc[j] =
            __builtin_amdgcn_mfma_f32_32x32x16_fp8_fp8(one8, one8, acc[j], 0, 0, 0);
      }
    }
  }
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < kAccums; ++j) s += acc[j][0];
  out[blockIdx.x * blockDim.x + threadIdx.x] = s;
}

template <typename Kernel>
py::dict run_mfma_stress(Kernel kernel, int iters, int workgroups,
                         const char* dtype) {
  if (iters <= 0 || iters > (1 << 17)) throw std::invalid_argument("iters");
  const int threads = 256;  // 4 waves per workgroup
  float* d_out = nullptr;
  const size_t out_elems = (size_t)workgroups * threads;
  HIP_CHECK(hipMalloc(&d_out, out_elems * sizeof(float)));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  // one warmup launch
  hipLaunchKernelGGL(kernel, dim3(workgroups), dim3(threads), 0, 0, d_out, 16);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  hipLaunchKernelGGL(kernel, dim3(workgroups), dim3(threads), 0, 0, d_out,
                     iters);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  // verify: each lane wrote sum of kAccums accumulator elements, each
  // exactly 16 * iters * kInnerUnroll (f32-exact while < 2^24)
  std::vector<float> host(out_elems);
  HIP_CHECK(hipMemcpy(host.data(), d_out, out_elems * sizeof(float),
                      hipMemcpyDeviceToHost));
  const double expect = (double)kAccums * 16.0 * iters * kInnerUnroll;
  size_t bad = 0;
  for (size_t i = 0; i < out_elems; ++i) {
    if (host[i] != (float)expect) bad++;
  }
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  // FLOPs: waves * mfma_count * 2*M*N*K
  const double waves = (double)workgroups * threads / 64.0;
  const double mfmas = waves * (double)iters * kInnerUnroll * kAccums;
  const double flops = mfmas * 2.0 * 32 * 32 * 16;
  py::dict d;
  d["dtype"] = dtype;
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds"] = ms * 1e-3;
  d["flops"] = flops;
  d["workgroups"] = workgroups;
  d["iters"] = iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict mfma_stress_bf16(int iters, int workgroups) {
  return run_mfma_stress(mfma_stress_bf16_kernel, iters, workgroups, "bf16");
}
py::dict mfma_stress_fp8(int iters, int workgroups) {
  return run_mfma_stress(mfma_stress_fp8_kernel, iters, workgroups, "fp8_e4m3");
}

// ---------------------------------------------------------------------------
// HBM bandwidth
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void hbm_triad_kernel(
    const float4* __restrict__ a, const float4* __restrict__ b,
    float4* __restrict__ c, size_t n, float s) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float4 av = a[i], bv = b[i];
    c[i] = make_float4(av.x + s * bv.x, av.y + s * bv.y, av.z + s * bv.z,
                       av.w + s * bv.w);
  }
}

__global__ __launch_bounds__(256) void hbm_read_kernel(
    const float4* __restrict__ a, float* __restrict__ out, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  float acc = 0.f;
  for (; i < n; i += stride) {
    float4 v = a[i];
    acc += v.x + v.y + v.z + v.w;
  }
  if (acc == -1.0f) out[0] = acc;  // never true; keeps the loads alive
}

py::dict hbm_bandwidth(double buffer_gb, int iters) {
  if (buffer_gb <= 0 || buffer_gb > 64) throw std::invalid_argument("buffer_gb");
  if (iters <= 0 || iters > 1000) throw std::invalid_argument("iters");
  const size_t n = (size_t)(buffer_gb * 1e9) / sizeof(float4);
  float4 *d_a = nullptr, *d_b = nullptr, *d_c = nullptr;
  float* d_sink = nullptr;
  HIP_CHECK(hipMalloc(&d_a, n * sizeof(float4)));
  HIP_CHECK(hipMalloc(&d_b, n * sizeof(float4)));
  HIP_CHECK(hipMalloc(&d_c, n * sizeof(float4)));
  HIP_CHECK(hipMalloc(&d_sink, sizeof(float)));
  HIP_CHECK(hipMemset(d_a, 0x3f, n * sizeof(float4)));
  HIP_CHECK(hipMemset(d_b, 0x3f, n * sizeof(float4)));
  const int threads = 256;
  const int blocks = 256 * 8;  // >> 256 CUs, fills all XCDs
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  // warmup
  hipLaunchKernelGGL(hbm_triad_kernel, dim3(blocks), dim3(threads), 0, 0, d_a,
                     d_b, d_c, n, 2.0f);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(hbm_triad_kernel, dim3(blocks), dim3(threads), 0, 0,
                       d_a, d_b, d_c, n, 2.0f);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float triad_ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&triad_ms, t0, t1));
  const double triad_bytes = (double)iters * 3.0 * n * sizeof(float4);

  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(hbm_read_kernel, dim3(blocks), dim3(threads), 0, 0, d_a,
                       d_sink, n);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float read_ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&read_ms, t0, t1));
  const double read_bytes = (double)iters * n * sizeof(float4);

  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_b));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipFree(d_sink));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  py::dict d;
  d["triad_gbps"] = triad_bytes / (triad_ms * 1e-3) / 1e9;
  d["read_gbps"] = read_bytes / (read_ms * 1e-3) / 1e9;
  d["buffer_bytes"] = (double)n * sizeof(float4);
  d["iters"] = iters;
  return d;
}

// ---------------------------------------------------------------------------
// LDS bandwidth
// ---------------------------------------------------------------------------

constexpr int kLdsFloat4 = 2048;  // 32 KiB of LDS per workgroup

__global__ __launch_bounds__(512, 2) void lds_bw_kernel(
    float* __restrict__ out, int iters) {
  __shared__ float4 buf[kLdsFloat4];
  const int tid = threadIdx.x;
  for (int i = tid; i < kLdsFloat4; i += blockDim.x) {
    buf[i] = make_float4(1.f, 2.f, 3.f, 4.f);
  }
  __syncthreads();
  float acc = 0.f;
  int idx = tid;
  for (int it = 0; it < iters; ++it) {
#pragma unroll 16
    for (int u = 0; u < 16; ++u) {
      // conflict-free wave64 ds_read_b128: consecutive lanes, consecutive
      // 16-B slots
      float4 v = buf[(idx + u * 512) & (kLdsFloat4 - 1)];
      acc += v.x + v.y + v.z + v.w;
    }
    idx += 17;  // rotate start to defeat trivial hoisting
  }
  if (acc == -1.0f) out[blockIdx.x] = acc;
}

py::dict lds_bandwidth(int iters, int workgroups) {
  if (iters <= 0 || iters > (1 << 22)) throw std::invalid_argument("iters");
  const int threads = 512;  // 8 waves/CU at 2 blocks/CU
  float* d_out = nullptr;
  HIP_CHECK(hipMalloc(&d_out, workgroups * sizeof(float)));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(lds_bw_kernel, dim3(workgroups), dim3(threads), 0, 0,
                     d_out, 16);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  hipLaunchKernelGGL(lds_bw_kernel, dim3(workgroups), dim3(threads), 0, 0,
                     d_out, iters);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double bytes =
      (double)workgroups * threads * (double)iters * 16.0 * sizeof(float4);
  py::dict d;
  d["lds_tbps"] = bytes / (ms * 1e-3) / 1e12;
  d["seconds"] = ms * 1e-3;
  d["workgroups"] = workgroups;
  return d;
}

// ---------------------------------------------------------------------------
// device info helper
// ---------------------------------------------------------------------------

py::dict device_info() {
  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  hipDeviceProp_t prop;
  py::dict d;
  d["device_count"] = ndev;
  if (ndev > 0) {
    HIP_CHECK(hipGetDeviceProperties(&prop, 0));
    d["name"] = std::string(prop.name);
    d["gcn_arch"] = std::string(prop.gcnArchName);
    d["multi_processor_count"] = prop.multiProcessorCount;
    d["total_mem_bytes"] = (double)prop.totalGlobalMem;
    d["clock_rate_khz"] = prop.clockRate;
  }
  return d;
}

void set_device(int dev) { HIP_CHECK(hipSetDevice(dev)); }

}  // namespace

PYBIND11_MODULE(_diag, m) {
  m.doc() = "gpud_amd CDNA4 diagnostic stress kernels (MFMA / HBM / LDS)";
  m.def("mfma_stress_bf16", &mfma_stress_bf16, py::arg("iters") = 4096,
        py::arg("workgroups") = 1024,
        "Register-resident bf16 MFMA stress; returns TFLOPS + verification");
  m.def("mfma_stress_fp8", &mfma_stress_fp8, py::arg("iters") = 4096,
        py::arg("workgroups") = 1024,
        "Register-resident fp8(e4m3) MFMA stress (non-scaled, bf16 rate)");
  m.def("hbm_bandwidth", &hbm_bandwidth, py::arg("buffer_gb") = 4.0,
        py::arg("iters") = 10, "float4 streaming triad + read over HBM3E");
  m.def("lds_bandwidth", &lds_bandwidth, py::arg("iters") = 100000,
        py::arg("workgroups") = 512, "ds_read_b128 LDS sweep");
  m.def("device_info", &device_info);
  m.def("set_device", &set_device, py::arg("device"));
}
