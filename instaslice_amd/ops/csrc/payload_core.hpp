// gfx950 HIP payload kernels: the workloads InstaSlice-AMD runs *inside*
// partitions to validate and benchmark them.
//
// The reference's only GPU workload is a prebuilt cuda-vectoradd container
// (samples/test-pod.yaml:12) used to prove a MIG slice works. Here the
// payloads are first-party CDNA4 kernels with a partition-verification angle:
//
//   vecadd       correctness smoke (the cuda-vectoradd analog)
//   membw        streaming-copy bandwidth probe: a CPX partition sees ~1/8 of
//                chip HBM bandwidth under NPS1 and its local quadrant's share
//                under NPS4 — measured, this *proves* the partition boundary
//   busy         bounded wall-clock spin filling the partition ("sleep pod")
//   xcd_census   every workgroup reports s_getreg_b32(HW_REG_XCC_ID): in a
//                CPX partition exactly one XCD may appear; in SPX all 8.
//                (XCC_ID read is a validation/performance tool, never a
//                correctness dependency — cdna_hip_programming.md §1.)
//
// Kernel style per the CDNA4 playbook: 256-thread blocks (4 waves of 64),
// float4 (dwordx4) streaming accesses for coalescing, grid-strided loops
// sized >> 256 workgroups to fill all 8 XCDs, bounded spins only.

#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <stdexcept>
#include <string>
#include <vector>

namespace payload {

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      throw std::runtime_error(std::string("HIP error: ") +               \
                               hipGetErrorString(_e) + " at " #expr);     \
    }                                                                     \
  } while (0)

constexpr int kBlock = 256;  // 4 wave64s per workgroup

// ---- kernels -------------------------------------------------------------

__global__ __launch_bounds__(kBlock) void vecadd_kernel(
    const float4* __restrict__ a, const float4* __restrict__ b,
    float4* __restrict__ c, size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < n4; i += stride) {
    float4 x = a[i], y = b[i];
    c[i] = make_float4(x.x + y.x, x.y + y.y, x.z + y.z, x.w + y.w);
  }
}

__global__ __launch_bounds__(kBlock) void fill_kernel(float4* __restrict__ p,
                                                      float v, size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  float4 f = make_float4(v, v, v, v);
  for (; i < n4; i += stride) p[i] = f;
}

// float4 streaming copy — the measured-bandwidth shape (6.29 TB/s whole-chip,
// MI355X_MICROARCH.md §Chip-level parameters).
__global__ __launch_bounds__(kBlock) void stream_copy_kernel(
    const float4* __restrict__ src, float4* __restrict__ dst, size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// Nontemporal variant: streamed data is used exactly once, so bypassing the
// L2/LLC write-allocate path frees cache bandwidth for the reads.
__global__ __launch_bounds__(kBlock) void stream_copy_nt_kernel(
    const float4* __restrict__ src, float4* __restrict__ dst, size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  // the builtin wants a scalar/vector-of-scalar pointer, not the
  // HIP_vector_type struct: reinterpret as the native 4-float vector
  typedef float vfloat4 __attribute__((ext_vector_type(4)));
  const vfloat4* s = reinterpret_cast<const vfloat4*>(src);
  vfloat4* d = reinterpret_cast<vfloat4*>(dst);
  for (; i < n4; i += stride) {
    vfloat4 v = __builtin_nontemporal_load(&s[i]);
    __builtin_nontemporal_store(v, &d[i]);
  }
}

// Bounded busy-wait on the constant-rate wall clock (s_memrealtime). The
// iteration guard bounds the spin even if the clock misbehaves
// (cdna_hip_programming.md §1: "bound every spin").
__global__ __launch_bounds__(kBlock) void busy_kernel(
    unsigned long long ticks, unsigned long long max_iters,
    unsigned int* __restrict__ sink) {
  unsigned long long start = wall_clock64();
  unsigned long long it = 0;
  unsigned int acc = 0;
  while (wall_clock64() - start < ticks && it < max_iters) {
    acc += static_cast<unsigned int>(it++);
  }
  if (threadIdx.x == 0 && acc == 0xFFFFFFFFu) *sink = acc;  // never taken
}

// One lane per workgroup records its XCD id. per_xcd must hold 8 counters.
__global__ __launch_bounds__(kBlock) void xcd_census_kernel(
    unsigned int* __restrict__ per_xcd) {
  if (threadIdx.x == 0) {
    unsigned int xcc;
    asm volatile("s_getreg_b32 %0, hwreg(HW_REG_XCC_ID)" : "=s"(xcc));
    atomicAdd(&per_xcd[xcc & 7], 1u);
  }
}

// ---- host wrappers -------------------------------------------------------

struct DeviceInfo {
  int device = 0;
  std::string name;
  std::string gcn_arch;
  int cu_count = 0;
  double total_mem_gb = 0;
  int xcd_count_visible = 0;  // from xcd census
};

inline int grid_for(size_t n4) {
  size_t blocks = (n4 + kBlock - 1) / kBlock;
  // >> 256 workgroups to fill 8 XCDs x 32 CUs (cap keeps launch sane)
  if (blocks > 65535) blocks = 65535;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

// Correctness smoke: c = a + b over n floats; returns max |err| (expect 0.0f:
// the sum 1.25 + 2.5 is exact in fp32).
inline double run_vecadd(size_t n, int device = 0) {
  HIP_CHECK(hipSetDevice(device));
  size_t n4 = (n + 3) / 4;
  float4 *a, *b, *c;
  HIP_CHECK(hipMalloc(&a, n4 * sizeof(float4)));
  HIP_CHECK(hipMalloc(&b, n4 * sizeof(float4)));
  HIP_CHECK(hipMalloc(&c, n4 * sizeof(float4)));
  int grid = grid_for(n4);
  hipLaunchKernelGGL(fill_kernel, dim3(grid), dim3(kBlock), 0, 0, a, 1.25f, n4);
  hipLaunchKernelGGL(fill_kernel, dim3(grid), dim3(kBlock), 0, 0, b, 2.5f, n4);
  hipLaunchKernelGGL(vecadd_kernel, dim3(grid), dim3(kBlock), 0, 0, a, b, c, n4);
  HIP_CHECK(hipGetLastError());
  std::vector<float4> host(n4);
  HIP_CHECK(hipMemcpy(host.data(), c, n4 * sizeof(float4), hipMemcpyDeviceToHost));
  double max_err = 0;
  for (const float4& v : host) {
    for (float f : {v.x, v.y, v.z, v.w}) {
      double e = static_cast<double>(f) - 3.75;
      if (e < 0) e = -e;
      if (e > max_err) max_err = e;
    }
  }
  HIP_CHECK(hipFree(a));
  HIP_CHECK(hipFree(b));
  HIP_CHECK(hipFree(c));
  return max_err;
}

// Streaming-copy bandwidth in GB/s (read+write bytes counted).
// blocks=0 picks the default grid; pass an explicit count to sweep the
// workgroups-per-CU space (MI355X: 256 CUs x 8 XCDs; the guide's rule is
// >> 256 workgroups to fill the chip).
inline double run_membw(size_t bytes, int iters, int device = 0, int blocks = 0,
                        bool nontemporal = false) {
  HIP_CHECK(hipSetDevice(device));
  size_t n4 = bytes / sizeof(float4);
  if (n4 == 0) throw std::invalid_argument("membw: bytes too small");
  float4 *src, *dst;
  HIP_CHECK(hipMalloc(&src, n4 * sizeof(float4)));
  HIP_CHECK(hipMalloc(&dst, n4 * sizeof(float4)));
  int grid = blocks > 0 ? blocks : grid_for(n4);
  auto kernel = nontemporal ? stream_copy_nt_kernel : stream_copy_kernel;
  hipLaunchKernelGGL(fill_kernel, dim3(grid), dim3(kBlock), 0, 0, src, 1.0f, n4);
  // warmup
  hipLaunchKernelGGL(kernel, dim3(grid), dim3(kBlock), 0, 0, src, dst, n4);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0, 0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(kernel, dim3(grid), dim3(kBlock), 0, 0, src, dst, n4);
  }
  HIP_CHECK(hipEventRecord(t1, 0));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  HIP_CHECK(hipFree(src));
  HIP_CHECK(hipFree(dst));
  double moved = 2.0 * static_cast<double>(n4) * sizeof(float4) * iters;
  return moved / (ms * 1e-3) / 1e9;
}

// Occupy the visible device for ~ms milliseconds (the "sleep pod" payload).
inline void run_busy(double ms, int device = 0) {
  HIP_CHECK(hipSetDevice(device));
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  // wall_clock64 rate: hipDeviceAttributeWallClockRate in kHz
  int khz = 0;
  HIP_CHECK(hipDeviceGetAttribute(&khz, hipDeviceAttributeWallClockRate, device));
  if (khz <= 0) khz = 100000;  // gfx9 constant clock default 100 MHz
  unsigned long long ticks =
      static_cast<unsigned long long>(ms * 1e-3 * khz * 1000.0);
  unsigned int* sink;
  HIP_CHECK(hipMalloc(&sink, sizeof(unsigned int)));
  int grid = prop.multiProcessorCount;  // one block per CU: full occupancy
  hipLaunchKernelGGL(busy_kernel, dim3(grid > 0 ? grid : 1), dim3(kBlock), 0, 0,
                     ticks, ~0ull, sink);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipFree(sink));
}

// Which XCDs executed our workgroups? Returns 8 counters.
inline std::vector<unsigned int> run_xcd_census(int device = 0, int blocks = 4096) {
  HIP_CHECK(hipSetDevice(device));
  unsigned int* per_xcd;
  HIP_CHECK(hipMalloc(&per_xcd, 8 * sizeof(unsigned int)));
  HIP_CHECK(hipMemset(per_xcd, 0, 8 * sizeof(unsigned int)));
  hipLaunchKernelGGL(xcd_census_kernel, dim3(blocks), dim3(kBlock), 0, 0, per_xcd);
  HIP_CHECK(hipGetLastError());
  std::vector<unsigned int> host(8);
  HIP_CHECK(hipMemcpy(host.data(), per_xcd, 8 * sizeof(unsigned int),
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(per_xcd));
  return host;
}

inline DeviceInfo get_device_info(int device = 0) {
  DeviceInfo info;
  info.device = device;
  HIP_CHECK(hipSetDevice(device));
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  info.name = prop.name;
  info.gcn_arch = prop.gcnArchName;
  info.cu_count = prop.multiProcessorCount;
  info.total_mem_gb = static_cast<double>(prop.totalGlobalMem) / (1024.0 * 1024.0 * 1024.0);
  auto census = run_xcd_census(device, 2048);
  int xcds = 0;
  for (unsigned int c : census)
    if (c) ++xcds;
  info.xcd_count_visible = xcds;
  return info;
}

inline int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

}  // namespace payload
