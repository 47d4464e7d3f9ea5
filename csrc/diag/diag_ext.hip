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
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <type_traits>
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
// MX-scaled fp8 stress — the only path to the ~5 PF dense fp8 peak on CDNA4
// (non-scaled fp8 runs at the bf16 rate; cdna_hip_programming.md §3/§4:
// mfma_scale_*_f8f6f4 with per-32-element E8M0 block scales, K=64 for the
// 32x32 shape). All-ones fp8 (0x38) with unit scales (E8M0 0x7F = 2^0)
// keeps the exact-verification property: each MFMA adds K=64 per element.
// ---------------------------------------------------------------------------

typedef int i32x8 __attribute__((ext_vector_type(8)));

__global__ __launch_bounds__(256, 4) void mfma_stress_mxfp8_kernel(
    float* __restrict__ out, int iters) {
  i32x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = 0x38383838;  // 4x fp8 e4m3 1.0
    b[i] = 0x38383838;
  }
  const int unit_scale = 0x7F7F7F7F;  // E8M0 exponent-bias 127 = 1.0
  f32x16 acc[kAccums] = {};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int u = 0; u < kInnerUnroll; ++u) {
#pragma unroll
      for (int j = 0; j < kAccums; ++j) {
        acc[j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a, b, acc[j], /*cbsz fp8*/ 0, /*blgp fp8*/ 0,
            /*opsel_a*/ 0, unit_scale, /*opsel_b*/ 0, unit_scale);
      }
    }
  }
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < kAccums; ++j) s += acc[j][0];
  out[blockIdx.x * blockDim.x + threadIdx.x] = s;
}

__global__ __launch_bounds__(256, 4) void mfma_stress_mxfp4_kernel(
    float* __restrict__ out, int iters) {
  // fp4 e2m1 1.0 = 0b0010 per nibble; cbsz/blgp = 4 selects the fp4 format
  // (the ~10 PF dense path; fp6 runs at the same rate on CDNA4)
  i32x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = 0x22222222;
    b[i] = 0x22222222;
  }
  const int unit_scale = 0x7F7F7F7F;
  f32x16 acc[kAccums] = {};
  for (int it = 0; it < iters; ++it) {
#pragma unroll
    for (int u = 0; u < kInnerUnroll; ++u) {
#pragma unroll
      for (int j = 0; j < kAccums; ++j) {
        acc[j] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            a, b, acc[j], /*cbsz fp4*/ 4, /*blgp fp4*/ 4,
            0, unit_scale, 0, unit_scale);
      }
    }
  }
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < kAccums; ++j) s += acc[j][0];
  out[blockIdx.x * blockDim.x + threadIdx.x] = s;
}

template <typename Kernel>
py::dict run_mx_stress(Kernel kernel, int iters, int workgroups,
                       const char* dtype) {
  if (iters <= 0 || iters > (1 << 17)) throw std::invalid_argument("iters");
  const int threads = 256;
  float* d_out = nullptr;
  const size_t out_elems = (size_t)workgroups * threads;
  HIP_CHECK(hipMalloc(&d_out, out_elems * sizeof(float)));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
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
  std::vector<float> host(out_elems);
  HIP_CHECK(hipMemcpy(host.data(), d_out, out_elems * sizeof(float),
                      hipMemcpyDeviceToHost));
  const double expect = (double)kAccums * 64.0 * iters * kInnerUnroll;
  size_t bad = 0;
  for (size_t i = 0; i < out_elems; ++i)
    if (host[i] != (float)expect) bad++;
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double waves = (double)workgroups * threads / 64.0;
  const double mfmas = waves * (double)iters * kInnerUnroll * kAccums;
  const double flops = mfmas * 2.0 * 32 * 32 * 64;
  py::dict d;
  d["dtype"] = dtype;
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds"] = ms * 1e-3;
  d["workgroups"] = workgroups;
  d["iters"] = iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict mfma_stress_mxfp4(int iters, int workgroups) {
  return run_mx_stress(mfma_stress_mxfp4_kernel, iters, workgroups,
                       "mxfp4_e2m1");
}

py::dict mfma_stress_mxfp8(int iters, int workgroups) {
  if (iters <= 0 || iters > (1 << 17)) throw std::invalid_argument("iters");
  const int threads = 256;
  float* d_out = nullptr;
  const size_t out_elems = (size_t)workgroups * threads;
  HIP_CHECK(hipMalloc(&d_out, out_elems * sizeof(float)));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(mfma_stress_mxfp8_kernel, dim3(workgroups), dim3(threads),
                     0, 0, d_out, 16);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  hipLaunchKernelGGL(mfma_stress_mxfp8_kernel, dim3(workgroups), dim3(threads),
                     0, 0, d_out, iters);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  std::vector<float> host(out_elems);
  HIP_CHECK(hipMemcpy(host.data(), d_out, out_elems * sizeof(float),
                      hipMemcpyDeviceToHost));
  const double expect = (double)kAccums * 64.0 * iters * kInnerUnroll;
  size_t bad = 0;
  for (size_t i = 0; i < out_elems; ++i)
    if (host[i] != (float)expect) bad++;
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double waves = (double)workgroups * threads / 64.0;
  const double mfmas = waves * (double)iters * kInnerUnroll * kAccums;
  const double flops = mfmas * 2.0 * 32 * 32 * 64;
  py::dict d;
  d["dtype"] = "mxfp8_e4m3";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds"] = ms * 1e-3;
  d["flops"] = flops;
  d["workgroups"] = workgroups;
  d["iters"] = iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

// ---------------------------------------------------------------------------
// GEMM-shaped stress: C[M,N] = A[M,K] · Bt[N,K]^T in bf16, LDS-tiled with
// direct global->LDS DMA (the cdna_hip_programming.md §5 step-3 recipe:
// 128x128 block tile, BK=64, double-buffered 16-byte global_load_lds).
// Stresses MFMA + LDS + HBM together (the register stress above isolates
// the matrix pipe). Asymmetric operands (A varies by row, B by column)
// with exactly-representable products catch row/col index bugs
// (guide §3: a symmetric B would hide a transposed C-write).
// ---------------------------------------------------------------------------

constexpr int BM = 128, BN = 128, BK = 64;

__device__ __forceinline__ float a_val(int i) {
  return 0.25f * ((i % 5) + 1);
}
__device__ __forceinline__ float b_val(int j) {
  return 0.125f * ((j % 7) + 1);
}

__global__ __launch_bounds__(256) void gemm_fill_kernel(
    __hip_bfloat16* __restrict__ A, __hip_bfloat16* __restrict__ Bt, int M,
    int N, int K) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t na = (size_t)M * K, nb = (size_t)N * K;
  for (size_t x = i; x < na; x += stride)
    A[x] = __hip_bfloat16(a_val((int)(x / K)));
  for (size_t x = i; x < nb; x += stride)
    Bt[x] = __hip_bfloat16(b_val((int)(x / K)));
}

__global__ __launch_bounds__(256, 2) void gemm_bf16_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  // ONE __shared__ object (guide §5 trap 4a): [2 buffers][A|B][128*64] bf16
  __shared__ __hip_bfloat16 lds[2][2][BM * BK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;     // 4 waves
  const int lane = tid & 63;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 wave grid of 64x64 tiles
  const int tiles_n = (N + BN - 1) / BN;
  // blockIdx -> tile map: plain row-major. An XCD row-band remap was
  // measured SLOWER here (931->842 TF @8192: a per-XCD row band's A slice
  // is 16 MB, far over the 4 MiB XCD L2, while row-major striping already
  // shares B tiles chip-wide through the 256 MiB L3), so the simple map
  // stays.
  const int brow = (blockIdx.x / tiles_n) * BM;
  const int bcol = (blockIdx.x % tiles_n) * BN;
  const int ntiles = K / BK;

  // glds staging: each wave DMAs 8 KiB: rows [wave*32, wave*32+32) of the
  // A tile and of the Bt tile; one 1 KiB instruction covers 8 rows
  // (lane l -> row l/8, 16 B at col (l%8)*8), LDS image row-major [128][64].
  // A 64-elem bf16 row is exactly 128 B, so a straight layout makes the
  // column-wise ds_read_b128 fragment reads hit one bank per 16-lane group
  // (the guide's "glds K-tile trap", up to 16-way conflict). Fix: XOR-swizzle
  // the 16-B slot index with the row parity — applied to the per-lane GLOBAL
  // source address (glds LDS writes stay lane-linear), and undone in the
  // fragment-read addressing below.
  auto stage = [&](int buf, int kt) {
    const int k0 = kt * BK;
    const __hip_bfloat16* gA = A + (size_t)(brow + wave * 32) * K + k0;
    const __hip_bfloat16* gB = Bt + (size_t)(bcol + wave * 32) * K + k0;
    __hip_bfloat16* lA = &lds[buf][0][wave * 32 * BK];
    __hip_bfloat16* lB = &lds[buf][1][wave * 32 * BK];
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int row = g * 8 + (lane >> 3);
      const int slot = (lane & 7) ^ (row & 7);  // pre-swizzled source slot
      const int col = slot * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(gA +
              (size_t)row * K + col),
          (__attribute__((address_space(3))) unsigned int*)(lA + g * 8 * BK),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(gB +
              (size_t)row * K + col),
          (__attribute__((address_space(3))) unsigned int*)(lB + g * 8 * BK),
          16, 0, 0);
    }
  };

  f32x16 acc[2][2] = {};
  stage(0, 0);
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < ntiles) stage((kt + 1) & 1, kt + 1);
    // compute on the current buffer: 4 k-substeps of 16; fragment slot =
    // (ks*2 + (lane>>5)) XOR (row&7), undoing the staged swizzle
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      const int kslot = ks * 2 + (lane >> 5);
      const int arow0 = wr * 64 + (lane & 31);
      const int bcol0 = wc * 64 + (lane & 31);
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int arow = arow0 + m * 32;
        af[m] = *(const bf16x8*)&lds[cur][0][arow * BK +
                                             (kslot ^ (arow & 7)) * 8];
        const int bcolr = bcol0 + m * 32;
        bf[m] = *(const bf16x8*)&lds[cur][1][bcolr * BK +
                                             (kslot ^ (bcolr & 7)) * 8];
      }
#pragma unroll
      for (int m = 0; m < 2; ++m) {
#pragma unroll
        for (int n = 0; n < 2; ++n) {
          acc[m][n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // hipcc emits vmcnt(0) here while glds is in flight
  }

  // epilogue: C/D layout for 32x32 MFMA (guide §3):
  // col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 2; ++n) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row =
            brow + wr * 64 + m * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int col = bcol + wc * 64 + n * 32 + (lane & 31);
        C[(size_t)row * N + col] = acc[m][n][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v2: the 256² 8-phase structure (cdna_hip_programming.md §5 "The 256²
// 8-phase template"): 256x256 tile, BK=64, 8 waves (2M×4N, 512 threads),
// mfma_f32_16x16x32_bf16, 128 KiB double-buffered LDS, st_16x32 swizzle
// (byte ^= ((byte>>9)&1)<<5 — kills the 8-way bank conflict of 128-B rows),
// raw s_barrier + s_setprio(1) around the MFMA burst, and K-tile-boundary
// vmcnt(0) placed where only the CURRENT tile's global_load_lds are
// outstanding (prefetches for the next tile are issued in phases 0-1, so
// the drain at the next boundary has had ≥2 phases of cover).
// ---------------------------------------------------------------------------

typedef f32x4 accfrag_t;

__device__ __forceinline__ int swz(int byte_off) {
  // st_16x32-style XOR swizzle extended for this kernel's read pattern:
  // bank-slot bits of a 16-B ds_read_b128 granule are byte bits 4..7; the
  // guide's bit5^=bit9 alone leaves row bit1 unused (measured exactly
  // 2-way, SQ_LDS_BANK_CONFLICT/IDX_ACTIVE = 50%), so bit6^=bit8 folds it
  // in too -> all 16 slots distinct within each 16-lane group.
  return byte_off ^ (((byte_off >> 9) & 1) << 5) ^
         (((byte_off >> 8) & 1) << 6);
}

template <bool SETPRIO, int BARRIER_MASK = 0xF, int PANEL = 0>
__global__ __launch_bounds__(512, 2) void gemm_bf16_8phase_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  // [2 buffers][A image 32 KiB | B image 32 KiB]
  __shared__ __hip_bfloat16 lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;          // 8 waves
  const int lane = tid & 63;
  const int wave_m = wave >> 2;       // 0..1 -> output rows [wave_m*128, +128)
  const int wave_n = wave & 3;        // 0..3 -> output cols [wave_n*64, +64)
  const int tiles_n = N / TN;
  int brow, bcol;
  if (PANEL == -1 && (M / TM) % 16 == 0 && tiles_n % 16 == 0) {
    // two-level XCD-aware supertiling: consecutive ids round-robin the 8
    // XCDs (private L2 each), so give each XCD a compact 8-row x 4-col
    // sub-block (12 tile-bands in its L2) while the 8 sub-blocks tile a
    // 16x16 global supertile (32 bands at the shared level vs 40 for a
    // row-major walk).
    const int sub = blockIdx.x & 7;           // = XCD under round-robin
    const int k = (blockIdx.x & 255) >> 3;    // 0..31 within sub-block
    const int st = blockIdx.x >> 8;           // supertile index
    const int st_cols = tiles_n / 16;
    const int ST_r = (st / st_cols) * 16, ST_c = (st % st_cols) * 16;
    brow = (ST_r + (sub & 1) * 8 + (k & 7)) * TM;
    bcol = (ST_c + (sub >> 1) * 4 + (k >> 3)) * TN;
  } else if (PANEL > 0 && tiles_n % PANEL == 0) {
    // L2-locality supertiling: consecutive blockIdx values walk a
    // PANEL-column x tiles_m panel column-major, so the ~256 concurrently
    // resident workgroups cover a near-square tile set (16 A-bands +
    // 16 B-bands of L2/HBM traffic instead of 8+32 for row-major walk
    // at 8192). The XCD round-robin dispatch takes consecutive ids.
    const int tiles_m = M / TM;
    const int per_panel = PANEL * tiles_m;
    const int panel = blockIdx.x / per_panel;
    const int rem = blockIdx.x % per_panel;
    brow = (rem / PANEL) * TM;
    bcol = (panel * PANEL + rem % PANEL) * TN;
  } else {
    brow = (blockIdx.x / tiles_n) * TM;
    bcol = (blockIdx.x % tiles_n) * TN;
  }
  const int ntiles = K / TK;

  // staging: 64 slots of 1 KiB cover [A image | B image]; wave w owns
  // slots [w*8, w*8+8) (waves 0-3 stage A rows, 4-7 stage B cols).
  // Lane l of slot s writes LDS byte D = s*1024 + l*16 (lane-linear glds);
  // the content belongs at logical offset L = swz(D) (XOR is an involution),
  // so the GLOBAL source address is pre-swizzled per lane.
  auto stage_slots = [&](int buf, int kt, int s0, int nslots) {
    const int k0 = kt * TK;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (j >= nslots) break;
      const int s = wave * 8 + s0 + j;
      const int img = s >> 5;           // 0 = A, 1 = B
      const int s_img = s & 31;
      const int D = s_img * 1024 + lane * 16;
      const int L = swz(D);
      const int row = L >> 7;           // row of the [256][64] bf16 image
      const int k = (L & 127) >> 1;
      const __hip_bfloat16* g =
          img == 0 ? A + (size_t)(brow + row) * K + k0 + k
                   : Bt + (size_t)(bcol + row) * K + k0 + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)(
              &lds[buf][img * TM * TK] + s_img * 512),
          16, 0, 0);
    }
  };

  // fragment read for mfma_f32_16x16x32_bf16: lane l holds row
  // (base + (l&15)), k = (l>>4)*8 + e — 16 B at the swizzled offset of
  // byte row*128 + kh*64 + (l>>4)*16
  auto read_a = [&](int buf, int rbase, int kh) -> bf16x8 {
    const int r = rbase + (lane & 15);
    const int off = swz(r * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][0] + off);
  };
  auto read_b = [&](int buf, int cbase, int kh) -> bf16x8 {
    const int c = cbase + (lane & 15);
    const int off = swz(c * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][TM * TK] + off);
  };

  accfrag_t acc[8][4] = {};  // 8 row-frags x 4 col-frags of 16x16
  bf16x8 bfrag[4][2];        // per-K-tile B fragments (reused by all phases)
  bf16x8 afrag[2][2][2];     // ping-pong: [phase&1][row-frag][k-half]

  // prologue: stage tile 0 (all 8 slots per wave)
  stage_slots(0, 0, 0, 4);
  stage_slots(0, 0, 4, 4);
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    const bool has_next = kt + 1 < ntiles;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // loads for this phase: B fragments + quadrant-0 A fragments land at
      // the head of the tile; later quadrants' A fragments were issued
      // during the PREVIOUS phase and have a whole MFMA burst of cover,
      // so the partial lgkmcnt(4) below waits only for in-flight
      // next-phase reads, not this phase's operands
      if (p == 0) {
#pragma unroll
        for (int c = 0; c < 4; ++c) {
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            bfrag[c][kh] = read_b(cur, wave_n * 64 + c * 16, kh);
        }
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            afrag[0][rr][kh] = read_a(cur, wave_m * 128 + rr * 16, kh);
        }
      }
      if (p < 3) {
        // pipeline: issue phase p+1's A fragments before this burst
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            afrag[(p + 1) & 1][rr][kh] =
                read_a(cur, wave_m * 128 + (p + 1) * 32 + rr * 16, kh);
        }
      }
      // prefetch next tile early: 4 slots in phase 0, 4 in phase 1
      // (all 8 in phase 0 congests it: measured 1154 -> 1001 TF @8192;
      // 2/phase leaves the last pair under-covered at the boundary drain)
      if (has_next && p < 2) stage_slots(nxt, kt + 1, p * 4, 4);
      // single barrier per phase (trailing only; the leading barrier of
      // the template measured -8%: 1119 -> 1203 TF @4096 without it)
      if (p < 3) {
        asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
      } else {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      }
      if (SETPRIO) __builtin_amdgcn_s_setprio(1);
      // kh outer: 8 independent MFMAs between accumulator reuses (the
      // dependent-accumulator latency exceeds the issue interval)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            acc[p * 2 + rr][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[p & 1][rr][kh], bfrag[c][kh], acc[p * 2 + rr][c], 0, 0,
                0);
          }
        }
      }
      if (SETPRIO) __builtin_amdgcn_s_setprio(0);
      if (p == 3) {
        // K-tile boundary: only tile kt+1's 8 glds are outstanding, and
        // they were issued >=2 phases ago — this drain is cheap and no
        // *later* prefetch gets caught by it (the step-3 ceiling trap)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // phase barriers are a lockstep throttle, not a correctness need:
      // staging targets the buffer the PREVIOUS tile read, so only the
      // p==3 boundary barrier orders cross-wave LDS reuse. BARRIER_MASK
      // selects which phases synchronize (bit p): 0xF = the template's
      // full lockstep, 0xA = half, 0x8 = boundary-only.
      if (BARRIER_MASK & (1 << p)) __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row =
            brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// MX-fp8 GEMM stress: the 8-phase structure at the block-scaled fp8 rate
// (mfma_scale_f32_16x16x128_f8f6f4, K=128 per instruction, ~2x the bf16
// matrix rate — cdna_hip_programming.md §3). Unit E8M0 scales keep the
// exact-verification property while exercising the full MX datapath with
// LDS-staged operands. Same LDS budget as the bf16 kernel: fp8 rows of
// 128 elements are again 128 B wide, so the staging slots, the swizzle
// and the banking analysis carry over unchanged.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void gemm_fill_fp8_kernel(
    unsigned char* __restrict__ A, unsigned char* __restrict__ Bt, int M,
    int N, int K) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t na = (size_t)M * K, nb = (size_t)N * K;
  for (size_t x = i; x < na; x += stride) {
    __hip_fp8_e4m3 v(a_val((int)(x / K)));
    A[x] = v.__x;
  }
  for (size_t x = i; x < nb; x += stride) {
    __hip_fp8_e4m3 v(b_val((int)(x / K)));
    Bt[x] = v.__x;
  }
}

template <int BARRIER_MASK = 0xF>
__global__ __launch_bounds__(512, 2) void gemm_mxfp8_8phase_kernel(
    const unsigned char* __restrict__ A, const unsigned char* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 128;  // fp8: 128-wide K tile
  __shared__ unsigned char lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;
  const int unit_scale = 0x7F7F7F7F;  // E8M0 1.0 per 32-elem block

  // staging: fp8 row of TK=128 elements is 128 B — identical slot scheme
  // to the bf16 kernel (64 x 1 KiB slots, 8 per wave), same swizzle
  auto stage_slots = [&](int buf, int kt, int s0, int nslots) {
    const int k0 = kt * TK;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (j >= nslots) break;
      const int s = wave * 8 + s0 + j;
      const int img = s >> 5;
      const int s_img = s & 31;
      const int D = s_img * 1024 + lane * 16;
      const int L = swz(D);
      const int row = L >> 7;
      const int k = L & 127;  // fp8: 1 B per element
      const unsigned char* g =
          img == 0 ? A + (size_t)(brow + row) * K + k0 + k
                   : Bt + (size_t)(bcol + row) * K + k0 + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)(
              &lds[buf][img * TM * TK] + s_img * 1024),
          16, 0, 0);
    }
  };

  // fragment: lane l holds row base+(l&15), k = (l>>4)*32 + e (32 B -> two
  // 16-B reads at swizzled offsets)
  auto read_frag = [&](int buf, int img, int rbase) -> i32x8 {
    const int r = rbase + (lane & 15);
    const unsigned char* base = &lds[buf][img * TM * TK];
    const int b0 = swz(r * 128 + (lane >> 4) * 32);
    const int b1 = swz(r * 128 + (lane >> 4) * 32 + 16);
    const int4 lo = *(const int4*)(base + b0);
    const int4 hi = *(const int4*)(base + b1);
    i32x8 v;
    v[0] = lo.x; v[1] = lo.y; v[2] = lo.z; v[3] = lo.w;
    v[4] = hi.x; v[5] = hi.y; v[6] = hi.z; v[7] = hi.w;
    return v;
  };

  accfrag_t acc[8][4] = {};
  i32x8 bfrag[4];
  i32x8 afrag[2][2];

  stage_slots(0, 0, 0, 4);
  stage_slots(0, 0, 4, 4);
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    const bool has_next = kt + 1 < ntiles;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      if (p == 0) {
#pragma unroll
        for (int c = 0; c < 4; ++c)
          bfrag[c] = read_frag(cur, 1, wave_n * 64 + c * 16);
#pragma unroll
        for (int rr = 0; rr < 2; ++rr)
          afrag[0][rr] = read_frag(cur, 0, wave_m * 128 + rr * 16);
      }
      if (p < 3) {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr)
          afrag[(p + 1) & 1][rr] =
              read_frag(cur, 0, wave_m * 128 + (p + 1) * 32 + rr * 16);
      }
      if (has_next && p < 2) stage_slots(nxt, kt + 1, p * 4, 4);
      if (p < 3) {
        asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
      } else {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          acc[p * 2 + rr][c] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
              afrag[p & 1][rr], bfrag[c], acc[p * 2 + rr][c],
              /*cbsz fp8*/ 0, /*blgp fp8*/ 0,
              /*opsel_a*/ 0, unit_scale, /*opsel_b*/ 0, unit_scale);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      if (p == 3) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // see the bf16 kernel: only the p==3 boundary barrier is required
      if (BARRIER_MASK & (1 << p)) __builtin_amdgcn_s_barrier();
    }
  }

#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row =
            brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}


// ---------------------------------------------------------------------------
// GEMM v3: quadrant-phase schedule — the template's deep pipeline. All 8
// waves compute ONE 128x128 C-quadrant per phase (Q00->Q01->Q11->Q10), so
// each phase needs at most one NEW operand half-tile; halves are staged in
// consumption order (A0,B0,B1,A1) one per phase and retired by a per-phase
// s_waitcnt vmcnt(4) — in steady state NOTHING ever drains to vmcnt(0):
//   phase p0: issue A0(t+1) -> wait retires B1(t)  (used p1)
//   phase p1: issue B0(t+1) -> wait retires A1(t)  (used p2)
//   phase p2: issue B1(t+1) -> wait retires A0(t+1) (early)
//   phase p3: issue A1(t+1) -> wait retires B0(t+1) (used t+1.p0)
// A retired half becomes cross-wave visible after that phase's trailing
// barrier, one phase before its first reader. Per-phase ds_reads follow
// the template's 12/4/8/4 pattern (A reloads at p0/p2, B at p0/p1/p3).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512, 2) void gemm_bf16_v3_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  __shared__ __hip_bfloat16 lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wm = wave >> 2;  // 0..1: 64-row band within the quadrant
  const int wn = wave & 3;   // 0..3: 32-col band within the quadrant
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  // stage one half (h: 0=A rows 0-127, 1=B rows 0-127, 2=B rows 128-255,
  // 3=A rows 128-255): 16 KiB = 16 slots, wave w issues slots {2w, 2w+1}
  auto stage_half = [&](int buf, int kt, int h) {
    const int img = (h == 0 || h == 3) ? 0 : 1;
    const int row0 = (h == 0 || h == 1) ? 0 : 128;
    const int k0 = kt * TK;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int s = wave * 2 + j;                  // slot within the half
      const int img_slot = (row0 >> 3) + s;        // slot within the image
      const int D = img_slot * 1024 + lane * 16;
      const int L = swz(D);
      const int row = L >> 7;
      const int k = (L & 127) >> 1;
      const __hip_bfloat16* g =
          img == 0 ? A + (size_t)(brow + row) * K + k0 + k
                   : Bt + (size_t)(bcol + row) * K + k0 + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)(
              &lds[buf][img * TM * TK] + img_slot * 512),
          16, 0, 0);
    }
  };

  auto read_a = [&](int buf, int rbase, int kh) -> bf16x8 {
    const int r = rbase + (lane & 15);
    const int off = swz(r * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][0] + off);
  };
  auto read_b = [&](int buf, int cbase, int kh) -> bf16x8 {
    const int c = cbase + (lane & 15);
    const int off = swz(c * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][TM * TK] + off);
  };

  // acc[p][rf][cf]: quadrant p, 4 row-frags x 2 col-frags of 16x16
  accfrag_t acc[4][4][2] = {};
  bf16x8 afrag[4][2];  // 4 row-frags x 2 k-halves (reloaded at p0/p2)
  bf16x8 bfrag[2][2];  // 2 col-frags x 2 k-halves (reloaded at p0/p1/p3)
  static constexpr int QM[4] = {0, 0, 1, 1};
  static constexpr int QN[4] = {0, 1, 1, 0};

  // prologue: A0+B0 of tile 0 fully landed; B1+A1 of tile 0 in flight
  stage_half(0, 0, 0);
  stage_half(0, 0, 1);
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();
  stage_half(0, 0, 2);
  stage_half(0, 0, 3);

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    const bool has_next = kt + 1 < ntiles;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      static constexpr int STAGE_H[4] = {0, 1, 2, 3};  // A0,B0,B1,A1
      if (has_next) stage_half(nxt, kt + 1, STAGE_H[p]);
      // retire schedule (see header comment); on the last tile drain the
      // remaining in-flight halves with counted waits instead
      if (has_next) {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      } else if (p == 0) {
        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      } else if (p == 1) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      // per-phase fragment loads (A at p0/p2, B at p0/p1/p3)
      if (p == 0 || p == 2) {
#pragma unroll
        for (int rf = 0; rf < 4; ++rf)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            afrag[rf][kh] =
                read_a(cur, QM[p] * 128 + wm * 64 + rf * 16, kh);
      }
      if (p != 2) {
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            bfrag[cf][kh] =
                read_b(cur, QN[p] * 128 + wn * 32 + cf * 16, kh);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
#pragma unroll
        for (int rf = 0; rf < 4; ++rf) {
#pragma unroll
          for (int cf = 0; cf < 2; ++cf) {
            acc[p][rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[rf][kh], bfrag[cf][kh], acc[p][rf][cf], 0, 0, 0);
          }
        }
      }
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

#pragma unroll
  for (int p = 0; p < 4; ++p) {
#pragma unroll
    for (int rf = 0; rf < 4; ++rf) {
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = brow + QM[p] * 128 + wm * 64 + rf * 16 +
                          (lane >> 4) * 4 + reg;
          const int col =
              bcol + QN[p] * 128 + wn * 32 + cf * 16 + (lane & 15);
          C[(size_t)row * N + col] = acc[p][rf][cf][reg];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v6: counted-wait chunk pipeline — the v2 glds structure with the
// K-tile boundary vmcnt(0) drain REMOVED. The 64 KiB tile is staged as 4
// chunks of 16 slots (C0 = B first-halves, C1 = A rows 0-63/128-191,
// C2 = B second-halves, C3 = A rows 64-127/192-255), one chunk per phase,
// every wave contributing 2 glds per chunk IN THE SAME ORDER — so after
// each wave waits s_waitcnt vmcnt(4) and passes a barrier, every chunk
// issued >=3 phases earlier is certified landed for ALL waves (vmcnt is
// per-wave; the uniform issue order + barrier extends it workgroup-wide).
// Phase p of a tile consumes (A half H=p>>1, B half C=p&1): 4x2x2 = 16
// MFMAs whose operands were staged 3-4 phases earlier — loads stay in
// flight across every barrier and nothing in the steady-state loop ever
// drains to vmcnt(0) (cdna_hip_programming.md §5 T3+T4: "counted-vs-drain0
// = +38%@4k / +73%@8k"; the v2 structure keeps a once-per-tile drain that
// its prefetch covers at 4096 but not at 8192 where loads come from HBM).
// The last tile keeps staging (wrapping to tile 0, never read) so the
// wait arithmetic stays uniform instead of under-waiting when the
// prefetch queue drains.
// ---------------------------------------------------------------------------

template <bool SETPRIO = true>
__global__ __launch_bounds__(512, 2) void gemm_bf16_v6_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  __shared__ __hip_bfloat16 lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  // chunk -> 16 image slots (1 KiB each, 8 rows of 128 B). Slot s of image
  // img covers rows [8s, 8s+8). B cols live as Bt rows.
  //   C0: B cols {64n+0..31}  = B slots {0-3, 8-11, 16-19, 24-27}
  //   C1: A rows 0-63,128-191 = A slots {0-7, 16-23}
  //   C2: B cols {64n+32..63} = B slots {4-7, 12-15, 20-23, 28-31}
  //   C3: A rows 64-127,192-255 = A slots {8-15, 24-31}
  // wave w stages slots chunk[2w], chunk[2w+1] — identical order per wave.
  auto chunk_slot = [&](int chunk, int j) -> int {
    const int k = wave * 2 + j;  // 0..15 within the chunk
    switch (chunk) {
      case 0: return (k >> 2) * 8 + (k & 3);          // B slots
      case 2: return (k >> 2) * 8 + 4 + (k & 3);      // B slots
      case 1: return (k >> 3) * 16 + (k & 7);         // A slots
      default: return (k >> 3) * 16 + 8 + (k & 7);    // A slots
    }
  };

  auto stage_chunk = [&](int buf, int kt, int chunk) {
    const int k0 = kt * TK;
    const int img = (chunk == 1 || chunk == 3) ? 0 : 1;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int s_img = chunk_slot(chunk, j);
      const int D = s_img * 1024 + lane * 16;
      const int L = swz(D);
      const int row = L >> 7;
      const int k = (L & 127) >> 1;
      const __hip_bfloat16* g =
          img == 0 ? A + (size_t)(brow + row) * K + k0 + k
                   : Bt + (size_t)(bcol + row) * K + k0 + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)(
              &lds[buf][img * TM * TK] + s_img * 512),
          16, 0, 0);
    }
  };

  auto read_a = [&](int buf, int rbase, int kh) -> bf16x8 {
    const int r = rbase + (lane & 15);
    const int off = swz(r * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][0] + off);
  };
  auto read_b = [&](int buf, int cbase, int kh) -> bf16x8 {
    const int c = cbase + (lane & 15);
    const int off = swz(c * 128 + kh * 64 + ((lane >> 4) * 16));
    return *(const bf16x8*)((const char*)&lds[buf][TM * TK] + off);
  };

  if (SETPRIO && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  accfrag_t acc[8][4] = {};
  bf16x8 afrag[4][2];  // phase-local: 4 row frags x 2 k-halves
  bf16x8 bfrag[2][2];  // phase-local: 2 col frags x 2 k-halves

  // prologue: tile 0 fully staged and drained once (the only vmcnt(0))
#pragma unroll
  for (int c = 0; c < 4; ++c) stage_chunk(0, 0, c);
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    // the last tile stages tile 0 again (into the buffer it no longer
    // reads) purely to keep every wave's outstanding-load count uniform
    const int kt_next = (kt + 1 < ntiles) ? kt + 1 : 0;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      stage_chunk(nxt, kt_next, p);
      // certify the chunk issued 3 phases ago: 2 glds per chunk, keep the
      // latest 2 chunks (4 loads) in flight — never 0 in the main loop
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      const int H = p >> 1;   // A half: rr 4H..4H+3
      const int Cc = p & 1;   // B half: c 2Cc..2Cc+1
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
        for (int kh = 0; kh < 2; ++kh)
          afrag[rr][kh] =
              read_a(cur, wave_m * 128 + H * 64 + rr * 16, kh);
      }
#pragma unroll
      for (int c = 0; c < 2; ++c) {
#pragma unroll
        for (int kh = 0; kh < 2; ++kh)
          bfrag[c][kh] = read_b(cur, wave_n * 64 + Cc * 32 + c * 16, kh);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
          for (int c = 0; c < 2; ++c) {
            acc[H * 4 + rr][Cc * 2 + c] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    afrag[rr][kh], bfrag[c][kh],
                    acc[H * 4 + rr][Cc * 2 + c], 0, 0, 0);
          }
        }
      }
    }
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v7: hand-scheduled asm K-loop on the v2 choreography — the guide's
// named lever past the plain-HIP plateau (cdna_hip_programming.md §5 "What
// does break it": a hand-scheduled K-loop with MFMA <-> load interleave and
// counted waits). Memory layout, swizzle, slot assignment, barrier and
// drain placement are byte-identical to v2; what changes is WHO schedules
// the phase body: each phase is one asm statement (generated by
// scripts/gen_v7_asm.py into gemm_v7_body.h) placing the next-phase
// ds_read_b128s and the next-tile global_load_lds pieces one-per-two-MFMAs
// inside the MFMA stream instead of hipcc's read-burst-then-MFMA-burst.
//
// Addressing exploits a provable property of the swizzle: in
//   x = r*128 + kh*64 + (lane>>4)*16,  r = base + q*32 + rr*16 + (lane&15)
// every term occupies disjoint bit ranges and bits 8..9 of x (the swz()
// XOR selectors) come from (lane&15)*128 alone — so swz(x) = x ^ mask(lane)
// and ONE per-lane base VGPR serves every fragment read of the kernel with
// 16-bit immediate offsets (A: q*4096+rr*2048+kh*64; B likewise off its
// own base). The glds pieces use the SADDR form (wave-uniform 64-bit SGPR
// base + per-lane 32-bit offset VGPR) so the slot walk is one v_add_u32.
// ---------------------------------------------------------------------------

#include "gemm_v7_body.h"

template <bool SETPRIO = false, int STYLE = 0>
__global__ __launch_bounds__(512, 2) void gemm_bf16_v7_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  __shared__ __hip_bfloat16 lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  // per-lane LDS fragment-read bases, one per MFMA k-half: kh*64 must live
  // INSIDE the swizzled part (the mask XORs bit 6; adding 64 across a set
  // mask bit would carry into bit 7)
  const uint32_t lds0 = (uint32_t)(uintptr_t)&lds[0][0];
  const int lane_raw = (lane & 15) * 128 + ((lane >> 4) * 16);
  const uint32_t swz_lane_k0 = (uint32_t)swz(lane_raw);
  const uint32_t swz_lane_k1 = (uint32_t)swz(lane_raw + 64);

  // glds geometry (v2 stage_slots equivalences): my wave owns slots
  // wave*8 .. wave*8+7 — image A for waves 0-3, image B for 4-7
  const int s_img0 = (wave * 8) & 31;
  const int glds_img = wave >> 2;
  const int D0 = s_img0 * 1024 + lane * 16;
  const int L0 = swz(D0);
  const int row0 = L0 >> 7;
  const int kcol0 = (L0 & 127) >> 1;
  const __hip_bfloat16* gptr =
      glds_img == 0 ? A + (size_t)brow * K : Bt + (size_t)bcol * K;
  uint64_t gbase;
  {
    const uint64_t p = (uint64_t)(uintptr_t)gptr;
    const uint32_t lo = __builtin_amdgcn_readfirstlane((uint32_t)p);
    const uint32_t hi = __builtin_amdgcn_readfirstlane((uint32_t)(p >> 32));
    gbase = ((uint64_t)hi << 32) | lo;
  }
  const uint32_t gstride =
      __builtin_amdgcn_readfirstlane((uint32_t)(8u * (uint32_t)K * 2u));
  const uint32_t voff_lane = ((uint32_t)row0 * (uint32_t)K + kcol0) * 2u;
  const uint32_t gdest_base = __builtin_amdgcn_readfirstlane(
      lds0 + (uint32_t)glds_img * 32768u + (uint32_t)s_img0 * 1024u);

  if (SETPRIO && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  accfrag_t acc[8][4] = {};
  bf16x8 afrag[2][2][2] = {};
  bf16x8 bfrag[4][2] = {};

  // prologue: stage tile 0 (all 8 of my slots) with the builtin glds —
  // identical addressing to the asm pieces, drained once
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int s_img = s_img0 + j;
    const int D = s_img * 1024 + lane * 16;
    const int L = swz(D);
    const int row = L >> 7;
    const int k = (L & 127) >> 1;
    const __hip_bfloat16* g = gptr + (size_t)row * K + k;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(
            &lds[0][glds_img * TM * TK] + s_img * 512),
        16, 0, 0);
  }
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    // last tile wraps its (never-read) prefetch to tile 0 so the wait
    // arithmetic stays uniform (same trick as v6)
    const int kt_next = (kt + 1 < ntiles) ? kt + 1 : 0;
    const uint32_t abase =
        lds0 + (uint32_t)cur * 65536u + (uint32_t)wave_m * 16384u;
    const uint32_t bbase = lds0 + (uint32_t)cur * 65536u + 32768u +
                           (uint32_t)wave_n * 8192u;
    uint32_t aaddr0 = abase + swz_lane_k0;
    uint32_t aaddr1 = abase + swz_lane_k1;
    uint32_t baddr0 = bbase + swz_lane_k0;
    uint32_t baddr1 = bbase + swz_lane_k1;
    uint32_t voff = voff_lane + (uint32_t)kt_next * 128u;
    uint32_t gdest = gdest_base + (uint32_t)nxt * 65536u;
    uint32_t mscratch;
    if constexpr (STYLE == 1) {
      // in-stream s_setprio(1..0) per phase (dynamic priority form)
      V7S_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V7S_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1, voff, gdest, mscratch,
                 gbase, gstride);
      V7S_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7S_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else if constexpr (STYLE == 2) {
      // memory groups placed late (under MFMAs 8-16)
      V7L_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V7L_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1, voff, gdest, mscratch,
                 gbase, gstride);
      V7L_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7L_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else if constexpr (STYLE == 3) {
      // E: reordered p0 head — kh-grouped reads, incremental counted waits
      V7E_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V7E_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1, voff, gdest, mscratch,
                 gbase, gstride);
      V7E_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7E_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else if constexpr (STYLE == 4) {
      // F: all 8 glds slots issued in phase 0 (max cover before the
      // boundary drain); phase 1 carries only its ds_read prefetches
      V7F_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V7F_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1);
      V7F_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7F_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else if constexpr (STYLE == 5) {
      // X = E + F combined
      V7X_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V7X_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1);
      V7X_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7X_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else if constexpr (STYLE == 6 || STYLE == 7) {
      // diagnostic skeletons (NOT health checks — wrong results by
      // construction, no verification): the X schedule with the glds
      // stream removed (6: barrier kept; 7: barrier removed too) to
      // attribute the residual wave-park between ds_read latency /
      // barrier skew / glds-boundary cover
      V7N_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1);
      V7N_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1);
      V7N_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7N_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    } else {
      V7_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                gdest, mscratch, gbase, gstride);
      V7_PHASE1(acc, afrag, bfrag, aaddr0, aaddr1, voff, gdest, mscratch,
                gbase, gstride);
      V7_PHASE2(acc, afrag, bfrag, aaddr0, aaddr1);
      V7_PHASE3(acc, afrag, bfrag, aaddr0, aaddr1);
    }
    // K-tile boundary: tile kt+1's 8 glds were issued in phases 0-1 with
    // 2-3 phases of MFMA cover (v2 semantics). The skeletons have no
    // glds in flight (6: barrier-only boundary; 7: free-running waves).
    if constexpr (STYLE < 6) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    } else if constexpr (STYLE == 6) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v7P: the X schedule with the phase-3 kh1 half CARRIED ACROSS THE
// BARRIER. Skeleton attribution (gemm_skel log) decomposed v7X's ~32%
// wave-park into ~145 TF barrier skew + ~290 TF glds boundary + ~400 TF
// schedule/read-latency; the biggest single exposure is the post-barrier
// phase-0 head, where every wave stalls on 6-12 LDS read returns with no
// ready work. Here phase 3 runs only its kh0 burst before the boundary;
// its kh1 burst (operands register-resident: afrag[1][..][1] + the OLD
// B-fragment set) executes AFTER the barrier, giving each wave 8 MFMAs
// (~128 XDL cycles) of latency-free work that covers the phase-0 read
// head. Costs: B fragments double-buffered per tile (+32 VGPR -> ~238)
// and the loop unrolls two K-tiles per iteration for the B-set
// ping-pong. First iteration's carried MFMAs multiply zero-initialised
// fragments (a 0*0 accumulate) so the pipeline needs no peel; the last
// tile's carried half runs in V7P_TAIL after the loop.
// ---------------------------------------------------------------------------

template <bool GFIRST = false, bool NTSTORE = false>
__global__ __launch_bounds__(512, 2) void gemm_bf16_v7p_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  __shared__ __hip_bfloat16 lds[2][2 * TM * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;
  const int wave_n = wave & 3;
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;  // multiple of 4 for every valid size

  const uint32_t lds0 = (uint32_t)(uintptr_t)&lds[0][0];
  const int lane_raw = (lane & 15) * 128 + ((lane >> 4) * 16);
  const uint32_t swz_lane_k0 = (uint32_t)swz(lane_raw);
  const uint32_t swz_lane_k1 = (uint32_t)swz(lane_raw + 64);

  const int s_img0 = (wave * 8) & 31;
  const int glds_img = wave >> 2;
  const int D0 = s_img0 * 1024 + lane * 16;
  const int L0 = swz(D0);
  const int row0 = L0 >> 7;
  const int kcol0 = (L0 & 127) >> 1;
  const __hip_bfloat16* gptr =
      glds_img == 0 ? A + (size_t)brow * K : Bt + (size_t)bcol * K;
  uint64_t gbase;
  {
    const uint64_t p = (uint64_t)(uintptr_t)gptr;
    const uint32_t lo = __builtin_amdgcn_readfirstlane((uint32_t)p);
    const uint32_t hi = __builtin_amdgcn_readfirstlane((uint32_t)(p >> 32));
    gbase = ((uint64_t)hi << 32) | lo;
  }
  const uint32_t gstride =
      __builtin_amdgcn_readfirstlane((uint32_t)(8u * (uint32_t)K * 2u));
  const uint32_t voff_lane = ((uint32_t)row0 * (uint32_t)K + kcol0) * 2u;
  const uint32_t gdest_base = __builtin_amdgcn_readfirstlane(
      lds0 + (uint32_t)glds_img * 32768u + (uint32_t)s_img0 * 1024u);

  accfrag_t acc[8][4] = {};
  bf16x8 afrag[2][2][2] = {};
  bf16x8 bfrag[2][4][2] = {};  // per-tile ping-pong (cross-barrier carry)

  // prologue: stage tile 0, drain once (the only vmcnt(0) with no cover)
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int s_img = s_img0 + j;
    const int D = s_img * 1024 + lane * 16;
    const int L = swz(D);
    const int row = L >> 7;
    const int k = (L & 127) >> 1;
    const __hip_bfloat16* g = gptr + (size_t)row * K + k;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(
            &lds[0][glds_img * TM * TK] + s_img * 512),
        16, 0, 0);
  }
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __builtin_amdgcn_s_barrier();

#define V7P_TILE(KT, BC, BO)                                                 \
  {                                                                          \
    const int kt = (KT);                                                     \
    const int cur = kt & 1;                                                  \
    const int nxt = cur ^ 1;                                                 \
    const int kt_next = (kt + 1 < ntiles) ? kt + 1 : 0;                      \
    const uint32_t abase =                                                   \
        lds0 + (uint32_t)cur * 65536u + (uint32_t)wave_m * 16384u;           \
    const uint32_t bbase = lds0 + (uint32_t)cur * 65536u + 32768u +          \
                           (uint32_t)wave_n * 8192u;                         \
    uint32_t aaddr0 = abase + swz_lane_k0;                                   \
    uint32_t aaddr1 = abase + swz_lane_k1;                                   \
    uint32_t baddr0 = bbase + swz_lane_k0;                                   \
    uint32_t baddr1 = bbase + swz_lane_k1;                                   \
    uint32_t voff = voff_lane + (uint32_t)kt_next * 128u;                    \
    uint32_t gdest = gdest_base + (uint32_t)nxt * 65536u;                    \
    uint32_t mscratch;                                                       \
    if constexpr (GFIRST)                                                    \
      V7P2_PHASE0(acc, afrag, bfrag[BC], bfrag[BO], aaddr0, aaddr1, baddr0,  \
                  baddr1, voff, gdest, mscratch, gbase, gstride);            \
    else                                                                     \
      V7P_PHASE0(acc, afrag, bfrag[BC], bfrag[BO], aaddr0, aaddr1, baddr0,   \
                 baddr1, voff, gdest, mscratch, gbase, gstride);             \
    V7X_PHASE1(acc, afrag, bfrag[BC], aaddr0, aaddr1);                       \
    V7X_PHASE2(acc, afrag, bfrag[BC], aaddr0, aaddr1);                       \
    V7P_PHASE3A(acc, afrag, bfrag[BC]);                                      \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                         \
    __builtin_amdgcn_s_barrier();                                            \
  }

  for (int kt = 0; kt < ntiles; kt += 2) {
    V7P_TILE(kt, 0, 1);
    V7P_TILE(kt + 1, 1, 0);
  }
#undef V7P_TILE
  // the final tile's carried kh1 half (its B set is bfrag[1]: ntiles even)
  V7P_TAIL(acc, afrag, bfrag[1]);

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg.
  // NTSTORE: C is written once and never re-read, but at 8192 its 256 MB
  // of store traffic evicts the A+B working set (exactly the 256 MB LLC)
  // — non-temporal stores keep the LLC for the operands.
#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        if constexpr (NTSTORE)
          __builtin_nontemporal_store(acc[R][c][reg],
                                      &C[(size_t)row * N + col]);
        else
          C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v8: 3-buffer glds ring — the guide's "glds with >1 tile in flight
// across the barrier" structure (cdna_hip_programming.md §5 "Pipelining
// across barriers": 2-buf overlap +40%, 3-buf span +83% vs serial at
// 1-block/CU occupancy and vgpr>=200 — exactly this kernel's regime).
// v7's 2 buffers cap glds cover at ONE tile (~1.2 us), marginal against
// loaded HBM latency; here every glds is issued TWO tiles before its
// data is read and the K-tile boundary is a counted per-wave
// s_waitcnt vmcnt(N) + lgkmcnt(0) + raw s_barrier — no vmcnt(0) drain
// anywhere in the steady state (and no __syncthreads(): its fence would
// emit vmcnt(0) and drain the in-flight glds this structure exists for).
// Tile 256x128x64 (48 KiB/buffer, 3 buffers = 144 KiB of the 160 KiB
// LDS); 8 waves as 4(M) x 2(N), wave tile 64x64, acc[4][4] = 64 VGPRs.
// The staging split keeps per-wave glds counts uniform for the counted
// wait: waves 0-3 stage the A image (8 slots/tile, boundary vmcnt(8)),
// waves 4-7 stage B (4 slots, vmcnt(4)). Two hand-scheduled asm phases
// per tile (one per MFMA k-half; bodies generated by
// scripts/gen_v8_asm.py): phase 0 reads all 16 fragments (kh0 batch
// first — lgkmcnt(8)) and carries ALL the tile's glds pieces (the v7
// E+F lessons); phase 1 is a pure MFMA burst. Addressing identities are
// v7's (swz() XOR mask is lane-only, so one swizzled per-lane base per
// k-half serves every fragment read with 16-bit immediates).
// ---------------------------------------------------------------------------

#include "gemm_v8_body.h"

__global__ __launch_bounds__(512, 2) void gemm_bf16_v8_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 128, TK = 64;
  constexpr int BUF = (TM + TN) * TK;  // bf16 elems: A image then B image
  __shared__ __hip_bfloat16 lds[3 * BUF];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 1;  // 0..3 — 64-row band
  const int wave_n = wave & 1;   // 0..1 — 64-col band
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  const uint32_t lds0 = (uint32_t)(uintptr_t)&lds[0];
  const int lane_raw = (lane & 15) * 128 + ((lane >> 4) * 16);
  const uint32_t swz_lane_k0 = (uint32_t)swz(lane_raw);
  const uint32_t swz_lane_k1 = (uint32_t)swz(lane_raw + 64);

  // staging geometry: A image = 32 slots (1 KiB each, 8 rows of 128 B),
  // B image = 16 slots at +32 KiB. Waves 0-3 stage A slots wave*8..+8,
  // waves 4-7 stage B slots (wave-4)*4..+4.
  const int glds_img = wave >> 2;  // 0 = A, 1 = B
  const int nglds = glds_img == 0 ? 8 : 4;
  const int s_img0 = glds_img == 0 ? wave * 8 : (wave - 4) * 4;
  const int D0 = s_img0 * 1024 + lane * 16;
  const int L0 = swz(D0);
  const int row0 = L0 >> 7;
  const int kcol0 = (L0 & 127) >> 1;
  const __hip_bfloat16* gptr =
      glds_img == 0 ? A + (size_t)brow * K : Bt + (size_t)bcol * K;
  uint64_t gbase;
  {
    const uint64_t p = (uint64_t)(uintptr_t)gptr;
    const uint32_t lo = __builtin_amdgcn_readfirstlane((uint32_t)p);
    const uint32_t hi = __builtin_amdgcn_readfirstlane((uint32_t)(p >> 32));
    gbase = ((uint64_t)hi << 32) | lo;
  }
  const uint32_t gstride =
      __builtin_amdgcn_readfirstlane((uint32_t)(8u * (uint32_t)K * 2u));
  const uint32_t voff_lane = ((uint32_t)row0 * (uint32_t)K + kcol0) * 2u;
  const uint32_t gdest_img = __builtin_amdgcn_readfirstlane(
      lds0 + (uint32_t)glds_img * (uint32_t)(TM * TK * 2) +
      (uint32_t)s_img0 * 1024u);  // + ring buffer offset at issue time

  accfrag_t acc[4][4] = {};
  bf16x8 afrag[4][2] = {};
  bf16x8 bfrag[4][2] = {};

  constexpr uint32_t BUFB = BUF * 2;  // bytes per ring buffer (49152)

  // prologue: stage tiles 0 and 1 into ring buffers 0,1 with the builtin
  // glds (identical addressing to the asm pieces), then a COUNTED wait —
  // tile 0 certified, tile 1 left in flight across the first barrier.
  for (int t = 0; t < 2; ++t) {
    for (int j = 0; j < nglds; ++j) {
      const int s_img = s_img0 + j;
      const int D = s_img * 1024 + lane * 16;
      const int L = swz(D);
      const int row = L >> 7;
      const int k = (L & 127) >> 1;
      const __hip_bfloat16* g = gptr + (size_t)row * K + (size_t)t * TK + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)(
              &lds[(size_t)t * BUF + glds_img * TM * TK] + s_img * 512),
          16, 0, 0);
    }
  }
  if (glds_img == 0)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  uint32_t cur_off = 0, nxt_off = BUFB, wr_off = 2 * BUFB;
  for (int kt = 0; kt < ntiles; ++kt) {
    // glds target: tile kt+2 into the retired ring buffer (dummy-wrapped
    // to tile 0 past the end — written, never read, keeps every wave's
    // outstanding-load count uniform)
    const int kt2 = (kt + 2 < ntiles) ? kt + 2 : 0;
    const uint32_t abase = lds0 + cur_off + (uint32_t)wave_m * 8192u;
    const uint32_t bbase =
        lds0 + cur_off + (uint32_t)(TM * TK * 2) + (uint32_t)wave_n * 8192u;
    uint32_t aaddr0 = abase + swz_lane_k0;
    uint32_t aaddr1 = abase + swz_lane_k1;
    uint32_t baddr0 = bbase + swz_lane_k0;
    uint32_t baddr1 = bbase + swz_lane_k1;
    uint32_t voff = voff_lane + (uint32_t)kt2 * (TK * 2);
    uint32_t gdest = gdest_img + wr_off;
    uint32_t mscratch;
    if (glds_img == 0) {
      V8A_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V8_PHASE1(acc, afrag, bfrag);
      // counted boundary: kt+1's 8 loads (issued LAST tile) certified,
      // kt+2's 8 stay in flight across the barrier
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      V8B_PHASE0(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff,
                 gdest, mscratch, gbase, gstride);
      V8_PHASE1(acc, afrag, bfrag);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const uint32_t rot = cur_off;
    cur_off = nxt_off;
    nxt_off = wr_off;
    wr_off = rot;
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 64 + rr * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[rr][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v9: TWO BLOCKS PER CU. The register-resident stress reaches 97.5%
// MfmaUtil because co-resident waves cover each other's stalls; v7P is
// capped at ~70% because its single 512-thread block (128 KiB LDS,
// 242 VGPR) owns the whole CU, so every barrier arrival and boundary
// drain parks all 8 waves together. v9 halves the tile to 128x128x64
// (32 KiB per LDS buffer, 64 KiB per block) and squeezes wave state to
// ~104 VGPRs — wave tile 64x32: acc[4][2] + 24 fragment regs —
// so two blocks co-reside (__launch_bounds__(512, 4) caps the
// allocation at 128 VGPRs): when one block drains/barriers, the other
// block's two waves per SIMD keep the XDL pipes fed. Costs measured
// into the A/B: 2x L2/LLC operand traffic (tile reuse halves) and
// double the boundary frequency per MFMA. LDS array headroom is known:
// v7 measures 25% SQ_LDS_IDX_ACTIVE, so 2x DS traffic stays far from
// the array limit. Body generated by scripts/gen_v9_asm.py (E+F
// schedule: kh-grouped reads, incremental lgkm waits, all 4 glds
// interleaved into the kh0 burst).
// ---------------------------------------------------------------------------

#include "gemm_v9_body.h"

__global__ __launch_bounds__(512, 4) void gemm_bf16_v9_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 128, TN = 128, TK = 64;
  __shared__ __hip_bfloat16 lds[2][(TM + TN) * TK];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;  // 0..1 — 64-row band
  const int wave_n = wave & 3;   // 0..3 — 32-col band
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  const uint32_t lds0 = (uint32_t)(uintptr_t)&lds[0][0];
  const int lane_raw = (lane & 15) * 128 + ((lane >> 4) * 16);
  const uint32_t swz_lane_k0 = (uint32_t)swz(lane_raw);
  const uint32_t swz_lane_k1 = (uint32_t)swz(lane_raw + 64);

  // staging: A image = 16 slots (1 KiB, 8 rows of 128 B), B image = 16
  // slots at +16 KiB. Waves 0-3 stage A slots wave*4..+4, waves 4-7
  // stage B slots (wave-4)*4..+4 — 4 glds per wave per tile, uniform.
  const int glds_img = wave >> 2;
  const int s_img0 = glds_img == 0 ? wave * 4 : (wave - 4) * 4;
  const int D0 = s_img0 * 1024 + lane * 16;
  const int L0 = swz(D0);
  const int row0 = L0 >> 7;
  const int kcol0 = (L0 & 127) >> 1;
  const __hip_bfloat16* gptr =
      glds_img == 0 ? A + (size_t)brow * K : Bt + (size_t)bcol * K;
  uint64_t gbase;
  {
    const uint64_t p = (uint64_t)(uintptr_t)gptr;
    const uint32_t lo = __builtin_amdgcn_readfirstlane((uint32_t)p);
    const uint32_t hi = __builtin_amdgcn_readfirstlane((uint32_t)(p >> 32));
    gbase = ((uint64_t)hi << 32) | lo;
  }
  const uint32_t gstride =
      __builtin_amdgcn_readfirstlane((uint32_t)(8u * (uint32_t)K * 2u));
  const uint32_t voff_lane = ((uint32_t)row0 * (uint32_t)K + kcol0) * 2u;
  const uint32_t gdest_base = __builtin_amdgcn_readfirstlane(
      lds0 + (uint32_t)glds_img * (uint32_t)(TM * TK * 2) +
      (uint32_t)s_img0 * 1024u);

  accfrag_t acc[4][2] = {};
  bf16x8 afrag[4][2] = {};
  bf16x8 bfrag[2][2] = {};

  // prologue: stage tile 0, drain once
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int s_img = s_img0 + j;
    const int D = s_img * 1024 + lane * 16;
    const int L = swz(D);
    const int row = L >> 7;
    const int k = (L & 127) >> 1;
    const __hip_bfloat16* g = gptr + (size_t)row * K + k;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(
            &lds[0][glds_img * TM * TK] + s_img * 512),
        16, 0, 0);
  }
  __builtin_amdgcn_s_waitcnt(0x3F70);  // vmcnt(0)
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    const int nxt = cur ^ 1;
    const int kt_next = (kt + 1 < ntiles) ? kt + 1 : 0;
    const uint32_t abase =
        lds0 + (uint32_t)cur * 32768u + (uint32_t)wave_m * 8192u;
    const uint32_t bbase = lds0 + (uint32_t)cur * 32768u + 16384u +
                           (uint32_t)wave_n * 4096u;
    uint32_t aaddr0 = abase + swz_lane_k0;
    uint32_t aaddr1 = abase + swz_lane_k1;
    uint32_t baddr0 = bbase + swz_lane_k0;
    uint32_t baddr1 = bbase + swz_lane_k1;
    uint32_t voff = voff_lane + (uint32_t)kt_next * 128u;
    uint32_t gdest = gdest_base + (uint32_t)nxt * 32768u;
    uint32_t mscratch;
    V9_TILE(acc, afrag, bfrag, aaddr0, aaddr1, baddr0, baddr1, voff, gdest,
            mscratch, gbase, gstride);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 64 + rr * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 32 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[rr][c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// GEMM v5: register-staged K-loop — NO LDS, NO barriers. Every wave loads
// its own MFMA fragments straight from global memory (both operands are
// K-contiguous, so a 16x16x32 fragment is 16 contiguous bytes per lane:
// lane l reads row base+(l&15), k-block (l>>4)*8 — no transpose, no LDS
// round trip), software-pipelined one quadrant / one B-tile ahead. With no
// cross-wave LDS coupling there is no s_barrier and no workgroup-level
// vmcnt drain anywhere in the K-loop — the structural stall that caps the
// glds templates (cdna_hip_programming.md §5 "the step-3 structure's
// ceiling": the path past the plain-HIP plateau is a K-loop whose prefetch
// loads stay in flight across tile boundaries, waits counted per wave and
// never drained to 0). Here hipcc itself emits the counted per-wave
// s_waitcnt vmcnt(N) at each fragment's first consumer because every load
// is an ordinary visible load (no glds in flight to force vmcnt(0) — §5
// "Three .s-level traps" (b) in reverse).
//
// Cross-wave reuse moves from LDS to the cache hierarchy: the 4 waves of a
// wave_m half re-read the same A rows through their CU's L1, and the 2
// waves of a wave_n band share B likewise; unique per-CU traffic per
// K-tile is the same 64 KiB a glds template stages. Register budget:
// acc 128 + bfrag 2x32 + afrag 2x16 = 224 VGPRs -> 2 waves/SIMD at
// __launch_bounds__(512, 2) (per-lane file 512; MI355X_MICROARCH.md
// register table).
// ---------------------------------------------------------------------------

template <bool SETPRIO = true>
__global__ __launch_bounds__(512, 2) void gemm_bf16_v5_kernel(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ Bt,
    float* __restrict__ C, int M, int N, int K) {
  constexpr int TM = 256, TN = 256, TK = 64;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wave_m = wave >> 2;  // 0..1 -> rows [wave_m*128, +128)
  const int wave_n = wave & 3;   // 0..3 -> cols [wave_n*64, +64)
  const int tiles_n = N / TN;
  const int brow = (blockIdx.x / tiles_n) * TM;
  const int bcol = (blockIdx.x % tiles_n) * TN;
  const int ntiles = K / TK;

  // per-lane fragment base: row (lane&15), k-block (lane>>4)*8 within the
  // 32-deep K half an mfma_f32_16x16x32_bf16 consumes
  const size_t lane_off = (size_t)(lane & 15) * K + (lane >> 4) * 8;
  const __hip_bfloat16* a0 = A + (size_t)(brow + wave_m * 128) * K + lane_off;
  const __hip_bfloat16* b0 = Bt + (size_t)(bcol + wave_n * 64) * K + lane_off;

  // static priority for the younger dispatch half (guide T5 static form:
  // per-burst setprio flips cost 3 spilled VGPRs here; the static form is
  // register-free). The condition must be wave-uniform via readfirstlane,
  // else s_setprio lands under an exec mask and applies to every wave.
  if (SETPRIO && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  f32x4 acc[8][4] = {};
  bf16x8 bfrag[2][4][2];  // [tile ping-pong][col group][k half]
  bf16x8 afrag[2][2][2];  // [quadrant ping-pong][row frag][k half]

  // NB: all array indices below are compile-time constants (unrolled loops
  // + the kt-by-2 buffer unroll) — a runtime buffer index would push these
  // ext_vector arrays to scratch (guide §5.4 rule 20).
  auto load_a = [&](int kt, int q, int buf) {
#pragma unroll
    for (int rr = 0; rr < 2; ++rr)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh)
        afrag[buf][rr][kh] = *(const bf16x8*)(
            a0 + (size_t)(q * 32 + rr * 16) * K + kt * TK + kh * 32);
  };
  auto load_b = [&](int kt, int c, int buf) {
#pragma unroll
    for (int kh = 0; kh < 2; ++kh)
      bfrag[buf][c][kh] = *(const bf16x8*)(
          b0 + (size_t)(c * 16) * K + kt * TK + kh * 32);
  };

  // prologue: tile 0's B fragments + quadrant 0, then enter steady state
#pragma unroll
  for (int c = 0; c < 4; ++c) load_b(0, c, 0);
  load_a(0, 0, 0);

  // one K-tile: 4 quadrant phases x 16 MFMAs; phase p prefetches quadrant
  // p+1 (next tile's q0 at p==3) and 2 of the next tile's 8 B fragments.
  // CUR/NXT are template-constant so the fragment arrays stay in registers.
  auto tile_body = [&](auto cur_c, int kt, bool has_next) {
    constexpr int CUR = decltype(cur_c)::value;
    constexpr int NXT = CUR ^ 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      if (p < 3) {
        load_a(kt, p + 1, (p + 1) & 1);
      } else if (has_next) {
        load_a(kt + 1, 0, 0);
      }
      if (has_next) load_b(kt + 1, p, NXT);
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            acc[p * 2 + rr][c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[p & 1][rr][kh], bfrag[CUR][c][kh], acc[p * 2 + rr][c],
                0, 0, 0);
          }
        }
      }
    }
  };

  // ntiles is even for every accepted size (size % 256 == 0 -> K/64 even)
  for (int kt = 0; kt < ntiles; kt += 2) {
    tile_body(std::integral_constant<int, 0>{}, kt, true);
    tile_body(std::integral_constant<int, 1>{}, kt + 1, kt + 2 < ntiles);
  }

  // epilogue: 16x16x32 C/D layout — col = lane&15, row = (lane>>4)*4 + reg
#pragma unroll
  for (int R = 0; R < 8; ++R) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = brow + wave_m * 128 + R * 16 + (lane >> 4) * 4 + reg;
        const int col = bcol + wave_n * 64 + c * 16 + (lane & 15);
        C[(size_t)row * N + col] = acc[R][c][reg];
      }
    }
  }
}

py::dict gemm_stress_bf16_v3(int size, int iters) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(gemm_bf16_v3_kernel, dim3(blocks), dim3(512), 0, 0, d_a,
                     d_bt, d_c, M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(gemm_bf16_v3_kernel, dim3(blocks), dim3(512), 0, 0,
                       d_a, d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s2 = 0; s2 < sample; ++s2)
      idx[s2] = ((size_t)s2 * 2654435761u) % ((size_t)M * N);
    for (int s2 = 0; s2 < sample; ++s2) {
      HIP_CHECK(hipMemcpy(&host[s2], d_c + idx[s2], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s2] / N), j = (int)(idx[s2] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s2] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256sq-quadrant-phase";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_mxfp8_impl(int size, int iters, int barrier_mask) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  unsigned char *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_fp8_kernel, dim3(2048), dim3(256), 0, 0, d_a,
                     d_bt, M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  auto* kern = barrier_mask == 0xF ? gemm_mxfp8_8phase_kernel<0xF>
                                   : gemm_mxfp8_8phase_kernel<0x8>;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0,
                     d_a, d_bt, d_c, M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0,
                       0, d_a, d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "mxfp8_e4m3";
  d["size"] = size;
  d["structure"] = "256sq-8phase";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v2_impl(int size, int iters, bool setprio,
                                  int barrier_mask = 0x8, int panel = 0) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  // boundary-only barriers shipped after a same-box interleaved A/B:
  // mask 0x8 vs 0xF measured 1301 vs 1193 TF @4096 and 1245 vs 1210 @8192
  // (3 reps each, <1% spread) — the per-phase lockstep costs ~9%/3%.
  auto* kern = setprio ? gemm_bf16_8phase_kernel<true, 0x8>
                       : gemm_bf16_8phase_kernel<false, 0x8>;
  if (barrier_mask == 0xA)
    kern = gemm_bf16_8phase_kernel<true, 0xA>;
  else if (barrier_mask == 0xF)
    kern = gemm_bf16_8phase_kernel<true, 0xF>;
  if (panel == -1) kern = gemm_bf16_8phase_kernel<true, 0x8, -1>;
  else if (panel == 8) kern = gemm_bf16_8phase_kernel<true, 0x8, 8>;
  else if (panel == 16) kern = gemm_bf16_8phase_kernel<true, 0x8, 16>;
  else if (panel == 32) kern = gemm_bf16_8phase_kernel<true, 0x8, 32>;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0,
                     d_a, d_bt, d_c, M, N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0,
                       d_a, d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256sq-8phase";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v2(int size, int iters) {
  return gemm_stress_bf16_v2_impl(size, iters, true);
}

py::dict gemm_stress_bf16_v5_impl(int size, int iters, bool setprio) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  auto* kern = setprio ? gemm_bf16_v5_kernel<true> : gemm_bf16_v5_kernel<false>;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c, M,
                     N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c,
                       M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256sq-regstage-nobarrier";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v7_impl(int size, int iters, bool setprio,
                                  int style = 0) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  auto* kern = setprio ? gemm_bf16_v7_kernel<true, 0>
                       : gemm_bf16_v7_kernel<false, 0>;
  if (style == 1) kern = gemm_bf16_v7_kernel<false, 1>;
  else if (style == 2) kern = gemm_bf16_v7_kernel<false, 2>;
  else if (style == 3) kern = gemm_bf16_v7_kernel<false, 3>;
  else if (style == 4) kern = gemm_bf16_v7_kernel<false, 4>;
  else if (style == 5) kern = gemm_bf16_v7_kernel<false, 5>;
  else if (style == 6) kern = gemm_bf16_v7_kernel<false, 6>;
  else if (style == 7) kern = gemm_bf16_v7_kernel<false, 7>;
  else if (style == 8) kern = gemm_bf16_v7p_kernel<false>;
  else if (style == 9) kern = gemm_bf16_v7p_kernel<true>;
  else if (style == 10) kern = gemm_bf16_v7p_kernel<false, true>;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c, M,
                     N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c,
                       M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  const bool skel = (style == 6 || style == 7);
  d["structure"] =
      skel ? "256sq-asm-kloop-SKELETON (wrong results by design)"
           : (style == 8 ? "256sq-asm-kloop-xbarrier-pipelined"
                         : "256sq-asm-kloop");
  d["skeleton"] = skel;
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v7(int size, int iters) {
  // Shipped default = style P (cross-barrier-pipelined phase 3 on the
  // E+F schedule), the 11-structure grid winner: ~1460-1490 TF @8192
  // (PMC MfmaUtil 70%), ~1390-1430 @4096 (gemm_ab_v7p/v7p2/v7p3 logs).
  // Rejected on A/B: in-burst + static setprio, late mem groups,
  // glds-burst-first head, nt C stores, 3-buffer ring (v8), counted
  // chunk pipeline (v6), register staging (v5).
  return gemm_stress_bf16_v7_impl(size, iters, false, 8);
}

py::dict gemm_stress_bf16_v7_style(int size, int iters, int style) {
  return gemm_stress_bf16_v7_impl(size, iters, false, style);
}

py::dict gemm_stress_bf16_v7_sp(int size, int iters) {
  return gemm_stress_bf16_v7_impl(size, iters, true);
}

py::dict gemm_stress_bf16_v7_nosp(int size, int iters) {
  return gemm_stress_bf16_v7_impl(size, iters, false);  // alias of default
}

py::dict gemm_stress_bf16_v7_graph(int size, int iters) {
  // v7P timed through ONE hipGraph of `iters` back-to-back launches —
  // removes the per-launch submission gaps from the timed region
  // (identical kernels and work; the guide's §6 launch-bound-loop lever)
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  auto* kern = gemm_bf16_v7p_kernel<false>;
  hipStream_t s;
  HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  HIP_CHECK(hipStreamBeginCapture(s, hipStreamCaptureModeGlobal));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, s, d_a, d_bt, d_c,
                       M, N, K);
  }
  hipGraph_t graph = nullptr;
  HIP_CHECK(hipStreamEndCapture(s, &graph));
  hipGraphExec_t ge = nullptr;
  HIP_CHECK(hipGraphInstantiate(&ge, graph, nullptr, nullptr, 0));
  HIP_CHECK(hipGraphLaunch(ge, s));  // warmup replay
  HIP_CHECK(hipStreamSynchronize(s));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0, s));
  HIP_CHECK(hipGraphLaunch(ge, s));
  HIP_CHECK(hipEventRecord(t1, s));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int ss = 0; ss < sample; ++ss)
      idx[ss] = ((size_t)ss * 2654435761u) % ((size_t)M * N);
    for (int ss = 0; ss < sample; ++ss) {
      HIP_CHECK(hipMemcpy(&host[ss], d_c + idx[ss], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[ss] / N), j = (int)(idx[ss] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[ss] != expect) bad++;
    }
  }
  HIP_CHECK(hipGraphExecDestroy(ge));
  HIP_CHECK(hipGraphDestroy(graph));
  HIP_CHECK(hipStreamDestroy(s));
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256sq-asm-kloop-xbarrier-pipelined-hipgraph";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v9(int size, int iters) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 128) * (N / 128);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(gemm_bf16_v9_kernel, dim3(blocks), dim3(512), 0, 0, d_a,
                     d_bt, d_c, M, N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(gemm_bf16_v9_kernel, dim3(blocks), dim3(512), 0, 0,
                       d_a, d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "128sq-2blocks-per-cu";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v8(int size, int iters) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 128);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(gemm_bf16_v8_kernel, dim3(blocks), dim3(512), 0, 0, d_a,
                     d_bt, d_c, M, N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(gemm_bf16_v8_kernel, dim3(blocks), dim3(512), 0, 0,
                       d_a, d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256x128-3buf-glds-ring";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v6_impl(int size, int iters, bool setprio) {
  if (size % 256 != 0 || size < 512 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 256 in [512,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / 256) * (N / 256);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  auto* kern = setprio ? gemm_bf16_v6_kernel<true> : gemm_bf16_v6_kernel<false>;
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c, M,
                     N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(512), 0, 0, d_a, d_bt, d_c,
                       M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  size_t bad = 0;
  {
    const int sample = 509;
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s)
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["structure"] = "256sq-chunk-counted-vmcnt";
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
}

py::dict gemm_stress_bf16_v6(int size, int iters) {
  return gemm_stress_bf16_v6_impl(size, iters, true);
}

py::dict gemm_stress_bf16_v6_nosp(int size, int iters) {
  return gemm_stress_bf16_v6_impl(size, iters, false);
}

py::dict gemm_stress_bf16_v5(int size, int iters) {
  return gemm_stress_bf16_v5_impl(size, iters, true);
}

py::dict gemm_stress_bf16_v5_nosp(int size, int iters) {
  return gemm_stress_bf16_v5_impl(size, iters, false);
}

// A/B seam: identical kernel with the s_setprio(1) MFMA-burst hint compiled
// out, for same-box interleaved comparisons (run-to-run and box-to-box DVFS
// variance exceeds the effect size, so cross-call comparisons are invalid).
py::dict gemm_stress_bf16_v2_nosp(int size, int iters) {
  return gemm_stress_bf16_v2_impl(size, iters, false);
}

py::dict gemm_stress_bf16_v2_bmask(int size, int iters, int barrier_mask) {
  return gemm_stress_bf16_v2_impl(size, iters, true, barrier_mask);
}

py::dict gemm_stress_bf16_v2_panel(int size, int iters, int panel) {
  return gemm_stress_bf16_v2_impl(size, iters, true, 0x8, panel);
}

py::dict gemm_stress_mxfp8(int size, int iters) {
  return gemm_stress_mxfp8_impl(size, iters, 0x8);
}

py::dict gemm_stress_mxfp8_bmask(int size, int iters, int barrier_mask) {
  return gemm_stress_mxfp8_impl(size, iters, barrier_mask);
}

py::dict gemm_stress_bf16(int size, int iters) {
  if (size % 128 != 0 || size < 256 || size > 16384)
    throw std::invalid_argument("size must be a multiple of 128 in [256,16384]");
  if (iters <= 0 || iters > 100) throw std::invalid_argument("iters");
  const int M = size, N = size, K = size;
  __hip_bfloat16 *d_a = nullptr, *d_bt = nullptr;
  float* d_c = nullptr;
  HIP_CHECK(hipMalloc(&d_a, (size_t)M * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_bt, (size_t)N * K * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&d_c, (size_t)M * N * sizeof(float)));
  hipLaunchKernelGGL(gemm_fill_kernel, dim3(2048), dim3(256), 0, 0, d_a, d_bt,
                     M, N, K);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  const int blocks = (M / BM) * (N / BN);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  hipLaunchKernelGGL(gemm_bf16_kernel, dim3(blocks), dim3(256), 0, 0, d_a,
                     d_bt, d_c, M, N, K);  // warmup
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(gemm_bf16_kernel, dim3(blocks), dim3(256), 0, 0, d_a,
                       d_bt, d_c, M, N, K);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  // verify a strided sample against the exact expected value
  size_t bad = 0;
  {
    const int sample = 257;  // co-prime stride walk
    std::vector<float> host(sample);
    std::vector<size_t> idx(sample);
    for (int s = 0; s < sample; ++s) {
      idx[s] = ((size_t)s * 2654435761u) % ((size_t)M * N);
    }
    for (int s = 0; s < sample; ++s) {
      HIP_CHECK(hipMemcpy(&host[s], d_c + idx[s], sizeof(float),
                          hipMemcpyDeviceToHost));
      const int i = (int)(idx[s] / N), j = (int)(idx[s] % N);
      const float expect =
          (float)K * (0.25f * ((i % 5) + 1)) * (0.125f * ((j % 7) + 1));
      if (host[s] != expect) bad++;
    }
  }
  HIP_CHECK(hipFree(d_a));
  HIP_CHECK(hipFree(d_bt));
  HIP_CHECK(hipFree(d_c));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  const double flops = (double)iters * 2.0 * M * (double)N * K;
  py::dict d;
  d["dtype"] = "bf16";
  d["size"] = size;
  d["tflops"] = flops / (ms * 1e-3) / 1e12;
  d["seconds_per_gemm"] = ms * 1e-3 / iters;
  d["verify_failures"] = (long)bad;
  d["verified"] = (bad == 0);
  return d;
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
  m.def("mfma_stress_mxfp4", &mfma_stress_mxfp4, py::arg("iters") = 2048,
        py::arg("workgroups") = 1024,
        "MX-scaled fp4 MFMA stress (the ~10 PF dense path)");
  m.def("mfma_stress_mxfp8", &mfma_stress_mxfp8, py::arg("iters") = 2048,
        py::arg("workgroups") = 1024,
        "MX-scaled fp8 MFMA stress (K=64 block-scaled, the ~5 PF dense path)");
  m.def("gemm_stress_bf16", &gemm_stress_bf16, py::arg("size") = 8192,
        py::arg("iters") = 5,
        "LDS-tiled bf16 GEMM stress (128x128 tile, BK=64, global_load_lds)");
  m.def("gemm_stress_bf16_v2", &gemm_stress_bf16_v2, py::arg("size") = 8192,
        py::arg("iters") = 5,
        "bf16 GEMM stress, 256^2 8-phase structure (swizzled LDS, setprio)");
  m.def("gemm_stress_bf16_v2_nosp", &gemm_stress_bf16_v2_nosp,
        py::arg("size") = 8192, py::arg("iters") = 5,
        "v2 GEMM with s_setprio disabled (A/B seam)");
  m.def("gemm_stress_bf16_v2_bmask", &gemm_stress_bf16_v2_bmask,
        py::arg("size") = 8192, py::arg("iters") = 5,
        py::arg("barrier_mask") = 0xF,
        "A/B variant of v2 with a phase-barrier mask (0xF/0xA/0x8)");
  m.def("gemm_stress_bf16_v2_panel", &gemm_stress_bf16_v2_panel,
        py::arg("size") = 8192, py::arg("iters") = 5, py::arg("panel") = 16,
        "A/B variant of v2 with L2 panel supertiling (8/16/32, 0=off)");
  m.def("gemm_stress_bf16_v7", &gemm_stress_bf16_v7, py::arg("size") = 8192,
        py::arg("iters") = 8,
        "hand-scheduled asm K-loop bf16 GEMM stress (v7)");
  m.def("gemm_stress_bf16_v7_graph", &gemm_stress_bf16_v7_graph,
        py::arg("size") = 8192, py::arg("iters") = 8,
        "v7P timed through one hipGraph of iters launches (no submission "
        "gaps in the timed region)");
  m.def("gemm_stress_bf16_v9", &gemm_stress_bf16_v9, py::arg("size") = 8192,
        py::arg("iters") = 8,
        "2-blocks-per-CU bf16 GEMM stress (v9): 128sq tile, cross-block "
        "stall cover");
  m.def("gemm_stress_bf16_v8", &gemm_stress_bf16_v8, py::arg("size") = 8192,
        py::arg("iters") = 8,
        "3-buffer glds-ring bf16 GEMM stress (v8): counted vmcnt boundary, "
        "raw barrier, >1 tile in flight");
  m.def("gemm_stress_bf16_v7_style", &gemm_stress_bf16_v7_style,
        py::arg("size") = 8192, py::arg("iters") = 8, py::arg("style") = 0,
        "v7 schedule-style seam: 0=base 1=in-burst setprio 2=late mem "
        "groups 3=reordered-p0-head 4=all-glds-p0 5=3+4 "
        "6=skeleton-no-glds 7=skeleton-no-glds-no-barrier "
        "8=cross-barrier-pipelined (P)");
  m.def("gemm_stress_bf16_v7_sp", &gemm_stress_bf16_v7_sp,
        py::arg("size") = 8192, py::arg("iters") = 8,
        "v7 with static young-half setprio (A/B seam)");
  m.def("gemm_stress_bf16_v7_nosp", &gemm_stress_bf16_v7_nosp,
        py::arg("size") = 8192, py::arg("iters") = 8,
        "v7 with the static setprio hint compiled out (A/B seam)");
  m.def("gemm_stress_bf16_v6", &gemm_stress_bf16_v6, py::arg("size") = 8192,
        py::arg("iters") = 8,
        "chunk-pipelined counted-vmcnt bf16 GEMM stress (v6, no boundary "
        "drain)");
  m.def("gemm_stress_bf16_v6_nosp", &gemm_stress_bf16_v6_nosp,
        py::arg("size") = 8192, py::arg("iters") = 8,
        "v6 with the static setprio hint compiled out (A/B seam)");
  m.def("gemm_stress_bf16_v5", &gemm_stress_bf16_v5, py::arg("size") = 8192,
        py::arg("iters") = 8,
        "register-staged no-LDS no-barrier bf16 GEMM stress (v5)");
  m.def("gemm_stress_bf16_v5_nosp", &gemm_stress_bf16_v5_nosp,
        py::arg("size") = 8192, py::arg("iters") = 8,
        "v5 with the s_setprio hint compiled out (A/B seam)");
  m.def("gemm_stress_bf16_v3", &gemm_stress_bf16_v3, py::arg("size") = 8192,
        py::arg("iters") = 5,
        "bf16 GEMM stress, quadrant-phase deep pipeline (no boundary drain)");
  m.def("gemm_stress_mxfp8_bmask", &gemm_stress_mxfp8_bmask,
        py::arg("size") = 8192, py::arg("iters") = 5,
        py::arg("barrier_mask") = 0xF,
        "A/B variant of the MX-fp8 GEMM with a phase-barrier mask");
  m.def("gemm_stress_mxfp8", &gemm_stress_mxfp8, py::arg("size") = 8192,
        py::arg("iters") = 5,
        "MX-scaled fp8 GEMM stress, 8-phase structure (K=128 MFMAs)");
  m.def("hbm_bandwidth", &hbm_bandwidth, py::arg("buffer_gb") = 4.0,
        py::arg("iters") = 10, "float4 streaming triad + read over HBM3E");
  m.def("lds_bandwidth", &lds_bandwidth, py::arg("iters") = 100000,
        py::arg("workgroups") = 512, "ds_read_b128 LDS sweep");
  m.def("device_info", &device_info);
  m.def("set_device", &set_device, py::arg("device"));
}
