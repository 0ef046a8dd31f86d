// cro_amd gfx950 device health probe.
//
// The MI355X-native replacement for the reference's "is the GPU alive"
// signal (forking `nvidia-smi` over pod exec, gpus.go:207-350): after the
// fabric composes a device and amdgpu binds it, this probe verifies the
// silicon actually computes before the operator advertises it via CDI:
//
//   1. mfma_f32_check  — one wave issues v_mfma_f32_16x16x4_f32 over a
//      K-loop; gfx950's f32-input MFMA is bitwise a k-ordered fmaf chain,
//      so the host can verify the result EXACTLY (no tolerance) — a
//      deterministic matrix-pipe + VGPR/AGPR integrity check.
//   2. bw_copy         — float4 streaming copy across a >>256-workgroup
//      grid, reporting achieved HBM3E bandwidth (expected ≈5-6.3 TB/s on a
//      healthy MI355X; gate is a loose floor).
//   3. mfma_bf16_rate  — dense v_mfma_f32_32x32x16_bf16 issue from 4
//      independent accumulators per wave; reports matrix-core TFLOP/s
//      (healthy ≈2.3-2.5 PF; loose floor gate).
//
// Built standalone with hipcc for gfx950 only (no torch dependency):
//   hipcc --offload-arch=gfx950 -O3 -fPIC -shared probe.hip -o libcroprobe.so
//
// C ABI consumed by cro_amd/nodeops/probe.py via ctypes.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <chrono>
#include <cstdio>
#include <cstring>

static double now_ms() {
  return std::chrono::duration<double, std::milli>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

#define CHECK(expr)                                                          \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      snprintf(out->msg, sizeof(out->msg), "%s failed: %s", #expr,           \
               hipGetErrorString(_e));                                       \
      out->ok = 0;                                                           \
      return -1;                                                             \
    }                                                                        \
  } while (0)

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

#include "croprobe.h"

// ---------------------------------------------------------------------------
// 1. exact f32 MFMA check (v_mfma_f32_16x16x4_f32: A 16x4, B 4x16, D 16x16;
//    lane l holds A[l&15][l>>4], B[l>>4][l&15]; D reg r -> row (l>>4)*4+r,
//    col l&15 — layouts per cdna_hip_programming.md §3)
// ---------------------------------------------------------------------------

__global__ void mfma_f32_check_kernel(const float* __restrict__ A,
                                      const float* __restrict__ B,
                                      float* __restrict__ D, int K) {
  int lane = threadIdx.x;  // exactly one wave of 64
  int row = lane & 15;
  int kh = lane >> 4;  // which of the 4 k slots this lane carries
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < K; k0 += 4) {
    float a = A[row * K + (k0 + kh)];
    float b = B[(k0 + kh) * 16 + row];  // col index == lane&15 == row bits
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  for (int r = 0; r < 4; ++r) {
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
  }
}

// ---------------------------------------------------------------------------
// 2. HBM streaming copy
// ---------------------------------------------------------------------------

__global__ void bw_copy_kernel(const float4* __restrict__ src,
                               float4* __restrict__ dst, size_t n4) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// ---------------------------------------------------------------------------
// 3. bf16 MFMA issue-rate (4 independent accumulators per wave keeps the
//    32-cycle/SIMD issue pipe of v_mfma_f32_32x32x16_bf16 saturated)
// ---------------------------------------------------------------------------

__global__ void mfma_bf16_rate_kernel(float* __restrict__ out, int iters) {
  bf16x8 a, b;
  // non-zero, non-uniform operands: zero inputs let DVFS overclock and
  // overstate the rate (MI355X_MICROARCH.md "DVFS give-back")
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)(0.5f + 0.0625f * (float)((threadIdx.x + i) & 7));
    b[i] = (__bf16)(0.25f + 0.03125f * (float)((threadIdx.x * 3 + i) & 7));
  }
  f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  for (int it = 0; it < iters; ++it) {
    acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
  }
  float s = 0.f;
  for (int i = 0; i < 16; ++i) s += acc0[i] + acc1[i] + acc2[i] + acc3[i];
  if (threadIdx.x == 0) out[blockIdx.x] = s;
}

// ---------------------------------------------------------------------------
// host driver
// ---------------------------------------------------------------------------

static void host_mfma_ref(const float* A, const float* B, float* D, int K) {
  // gfx950 f32-in MFMA is bitwise a k-ordered fmaf chain (guide §3):
  for (int r = 0; r < 16; ++r) {
    for (int c = 0; c < 16; ++c) {
      float acc = 0.f;
      for (int k = 0; k < K; ++k) acc = fmaf(A[r * K + k], B[k * 16 + c], acc);
      D[r * 16 + c] = acc;
    }
  }
}

// Per-device cached probe context: the probe runs on every attach, so
// allocations and events are created once and reused — only the kernels,
// copies and compares run per call.
struct ProbeCtx {
  int ready = 0;
  long long vram_total = 0, vram_free = 0;  // hipMemGetInfo is sysfs-backed
                                            // and ms-scale: snapshot at init
  float *dA = nullptr, *dB = nullptr, *dD = nullptr;
  float4 *src = nullptr, *dst = nullptr;
  float* sink = nullptr;
  hipEvent_t e0 = nullptr, e1 = nullptr;
  char gcn_arch[64] = {0};  // hipGetDeviceProperties costs milliseconds —
                            // queried once, the probe runs per attach
};
static ProbeCtx g_ctx[64];

extern "C" int cro_probe_run(int device, struct CroProbeResult* out) {
  memset(out, 0, sizeof(*out));
  out->ok = 0;
  double t0 = now_ms();

  int count = 0;
  CHECK(hipGetDeviceCount(&count));
  if (device < 0 || device >= count) {
    snprintf(out->msg, sizeof(out->msg), "device %d out of range (%d present)",
             device, count);
    return -1;
  }
  CHECK(hipSetDevice(device));

  // -- exact f32 MFMA ------------------------------------------------------
  out->t_setup_ms = now_ms() - t0;
  t0 = now_ms();
  const int K = 64;
  float hA[16 * K], hB[K * 16], hD[256], refD[256];
  unsigned s = 0x9e3779b9u;
  for (int i = 0; i < 16 * K; ++i) {
    s = s * 1664525u + 1013904223u;
    hA[i] = ((float)(s >> 8) / 16777216.0f) - 0.5f;
  }
  for (int i = 0; i < K * 16; ++i) {
    s = s * 1664525u + 1013904223u;
    hB[i] = ((float)(s >> 8) / 16777216.0f) - 0.5f;
  }
  ProbeCtx* ctx = (device < 64) ? &g_ctx[device] : nullptr;
  if (ctx == nullptr) {
    snprintf(out->msg, sizeof(out->msg), "device ordinal %d above cache limit", device);
    return -1;
  }
  const size_t bytes = (size_t)256 << 20;  // bw buffers (see below)
  if (!ctx->ready) {
    CHECK(hipMalloc(&ctx->dA, sizeof(hA)));
    CHECK(hipMalloc(&ctx->dB, sizeof(hB)));
    CHECK(hipMalloc(&ctx->dD, sizeof(hD)));
    CHECK(hipMalloc(&ctx->src, bytes));
    CHECK(hipMalloc(&ctx->dst, bytes));
    CHECK(hipMemset(ctx->src, 0x5a, bytes));
    CHECK(hipMalloc(&ctx->sink, 1024 * sizeof(float)));
    CHECK(hipEventCreate(&ctx->e0));
    CHECK(hipEventCreate(&ctx->e1));
    hipDeviceProp_t prop;
    CHECK(hipGetDeviceProperties(&prop, device));
    snprintf(ctx->gcn_arch, sizeof(ctx->gcn_arch), "%s", prop.gcnArchName);
    size_t free_b = 0, total_b = 0;
    CHECK(hipMemGetInfo(&free_b, &total_b));
    ctx->vram_total = (long long)total_b;
    ctx->vram_free = (long long)free_b;
    // one warm round at init covers kernel-code upload; per-call warms are
    // off the attach path
    hipLaunchKernelGGL(bw_copy_kernel, dim3(8192), dim3(256), 0, 0, ctx->src, ctx->dst,
                       bytes / sizeof(float4));
    hipLaunchKernelGGL(mfma_bf16_rate_kernel, dim3(1024), dim3(256), 0, 0, ctx->sink, 64);
    CHECK(hipDeviceSynchronize());
    ctx->ready = 1;
  }
  snprintf(out->gcn_arch, sizeof(out->gcn_arch), "%s", ctx->gcn_arch);
  out->vram_total = ctx->vram_total;
  out->vram_free = ctx->vram_free;
  CHECK(hipMemcpy(ctx->dA, hA, sizeof(hA), hipMemcpyHostToDevice));
  CHECK(hipMemcpy(ctx->dB, hB, sizeof(hB), hipMemcpyHostToDevice));
  hipLaunchKernelGGL(mfma_f32_check_kernel, dim3(1), dim3(64), 0, 0, ctx->dA, ctx->dB, ctx->dD, K);
  CHECK(hipGetLastError());
  CHECK(hipMemcpy(hD, ctx->dD, sizeof(hD), hipMemcpyDeviceToHost));
  host_mfma_ref(hA, hB, refD, K);
  out->mfma_f32_exact = (memcmp(hD, refD, sizeof(hD)) == 0) ? 1 : 0;

  out->t_mfma_ms = now_ms() - t0;
  t0 = now_ms();
  // -- HBM bandwidth -------------------------------------------------------
  // 256 MiB src + dst from the cached context (2 GiB of traffic per
  // measurement — ample signal without putting hipMalloc on the
  // attach-latency path or pinning more of the 288 GB than a gate needs)
  size_t n4 = bytes / sizeof(float4);
  hipEvent_t e0 = ctx->e0, e1 = ctx->e1;
  dim3 grid(8192), block(256);
  CHECK(hipEventRecord(e0));
  const int reps = 8;
  for (int i = 0; i < reps; ++i)
    hipLaunchKernelGGL(bw_copy_kernel, grid, block, 0, 0, ctx->src, ctx->dst, n4);
  CHECK(hipEventRecord(e1));
  CHECK(hipEventSynchronize(e1));
  float ms = 0.f;
  CHECK(hipEventElapsedTime(&ms, e0, e1));
  out->hbm_gbps = (double)(2.0 * bytes * reps) / (ms * 1e6);

  out->t_bw_ms = now_ms() - t0;
  t0 = now_ms();
  // -- bf16 MFMA rate ------------------------------------------------------
  const int blocks = 1024, iters = 2048;
  CHECK(hipEventRecord(e0));
  hipLaunchKernelGGL(mfma_bf16_rate_kernel, dim3(blocks), dim3(256), 0, 0, ctx->sink, iters);
  CHECK(hipEventRecord(e1));
  CHECK(hipEventSynchronize(e1));
  CHECK(hipEventElapsedTime(&ms, e0, e1));
  double waves = (double)blocks * 256.0 / 64.0;
  double flops = waves * 4.0 * (double)iters * 2.0 * 32.0 * 32.0 * 16.0;
  out->bf16_tflops = flops / (ms * 1e9);
  out->t_bf16_ms = now_ms() - t0;

  // gates: exact MFMA is hard; bandwidth/rate are loose floors so a busy or
  // power-capped chip never false-fails
  if (!out->mfma_f32_exact) {
    snprintf(out->msg, sizeof(out->msg), "f32 MFMA mismatch vs host fmaf chain");
    return -2;
  }
  if (out->hbm_gbps < 500.0) {
    snprintf(out->msg, sizeof(out->msg), "HBM bandwidth %.0f GB/s below floor", out->hbm_gbps);
    return -3;
  }
  if (out->bf16_tflops < 100.0) {
    snprintf(out->msg, sizeof(out->msg), "bf16 MFMA %.0f TF below floor", out->bf16_tflops);
    return -4;
  }
  out->ok = 1;
  snprintf(out->msg, sizeof(out->msg), "ok");
  return 0;
}

// Raw MFMA tile entry for numerics tests: D(16x16) = A(16xK) @ B(Kx16) on
// the f32 matrix pipe; compared against a PyTorch fp32 reference in
// tests/test_gpu.py.  K must be a multiple of 4.
extern "C" int cro_probe_mfma_f32(int device, const float* A, const float* B,
                                  float* D, int K) {
  struct CroProbeResult scratch;
  struct CroProbeResult* out = &scratch;  // reuse CHECK() plumbing
  memset(out, 0, sizeof(*out));
  if (K <= 0 || (K & 3) != 0) return -10;
  CHECK(hipSetDevice(device));
  float *dA, *dB, *dD;
  CHECK(hipMalloc(&dA, 16 * K * sizeof(float)));
  CHECK(hipMalloc(&dB, K * 16 * sizeof(float)));
  CHECK(hipMalloc(&dD, 256 * sizeof(float)));
  CHECK(hipMemcpy(dA, A, 16 * K * sizeof(float), hipMemcpyHostToDevice));
  CHECK(hipMemcpy(dB, B, K * 16 * sizeof(float), hipMemcpyHostToDevice));
  hipLaunchKernelGGL(mfma_f32_check_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD, K);
  CHECK(hipGetLastError());
  CHECK(hipMemcpy(D, dD, 256 * sizeof(float), hipMemcpyDeviceToHost));
  (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dD);
  return 0;
}

// Raw VRAM alloc/free for the self-identification fingerprint
// (cro_amd/nodeops/kfd.py resolve_self_kfd_pid): the operator finds its own
// host pid in /sys/class/kfd/kfd/proc by watching which vram_<gpu_id> file
// grows by a marker-sized hipMalloc.
extern "C" void* cro_probe_alloc(int device, long long bytes) {
  if (hipSetDevice(device) != hipSuccess) return nullptr;
  void* p = nullptr;
  if (hipMalloc(&p, (size_t)bytes) != hipSuccess) return nullptr;
  if (hipMemset(p, 1, (size_t)bytes) != hipSuccess) { (void)hipFree(p); return nullptr; }
  if (hipDeviceSynchronize() != hipSuccess) { (void)hipFree(p); return nullptr; }
  return p;
}

extern "C" void cro_probe_free(void* p) { (void)hipFree(p); }

extern "C" int cro_probe_device_count(void) {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess) return -1;
  return count;
}

extern "C" int cro_probe_pci_bus_id(int device, char* buf, int len) {
  if (hipDeviceGetPCIBusId(buf, len, device) != hipSuccess) return -1;
  return 0;
}
