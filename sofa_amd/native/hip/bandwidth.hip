// sofa-bandwidth — HIP bandwidth test workload for MI355X.
//
// The MI355X stand-in for the CUDA `bandwidthTest` the reference's BASELINE
// config 2 profiles (BASELINE.json: "hip bandwidthTest H2D/D2D on 1 MI355X").
// Measures H2D / D2H (pinned + pageable) via hipMemcpyAsync and device-local
// bandwidth via both hipMemcpyDtoD and a grid-striding vectorized copy kernel
// (uint4 = 16 B/lane, coalesced; HBM3E peak is 8 TB/s, ~6.3 achievable).
// Prints one CSV row per (kind, size) to stdout — the profiled artifact.
//
// `--ring LOGDIR`: the device-trace-ring product path.  The copy kernel is
// replaced by an instrumented variant where wave 0 of every workgroup brackets
// its copy loop with s_memrealtime and ring_pushes a per-workgroup span —
// intra-kernel events no host-side profiler can see (rocprofiler reports one
// span for the whole grid).  The ring is compacted on-device and dumped as an
// SGT file into LOGDIR, so `sofa stat "sofa-bandwidth 0 --ring <logdir>"`
// shows per-workgroup phases on the same timeline as the rocprofiler kernels
// (the XCD-imbalance view: 4096 workgroups over 256 CUs / 8 XCDs).

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>

#include "trace_ring.h"

// from libsofahip (linked; see native/build.py)
extern "C" int sofa_ring_create(int device, uint32_t capacity, void** ring_out);
extern "C" int sofa_ring_destroy(void* ring_p);
extern "C" int sofa_ring_device_ptrs(void* ring_p, void** ctl, void** slots);
extern "C" int sofa_ring_dump_sgt(void* ring_p, int device, const char* logdir,
                                  const char** tag_names, int n_tags);

#define HIP_CHECK(x)                                                          \
  do {                                                                        \
    hipError_t err_ = (x);                                                    \
    if (err_ != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(err_),     \
              __FILE__, __LINE__);                                            \
      exit(1);                                                                \
    }                                                                         \
  } while (0)

__global__ void copy_kernel(const uint4* __restrict__ src,
                            uint4* __restrict__ dst, size_t n_vec) {
  size_t i = (size_t) blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t) gridDim.x * blockDim.x;
  for (; i < n_vec; i += stride) dst[i] = src[i];
}

// ring-instrumented variant: per-workgroup span records (tag 1 = wg_copy,
// src = blockIdx, arg = bytes this workgroup moved)
__global__ void copy_kernel_ring(const uint4* __restrict__ src,
                                 uint4* __restrict__ dst, size_t n_vec,
                                 RingControl* ctl, RingRec* slots) {
  uint64_t t0 = 0;
  if (threadIdx.x == 0) t0 = __builtin_amdgcn_s_memrealtime();
  size_t i = (size_t) blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t) gridDim.x * blockDim.x;
  size_t moved = 0;
  for (; i < n_vec; i += stride) {
    dst[i] = src[i];
    moved += sizeof(uint4);
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t t1 = __builtin_amdgcn_s_memrealtime();
    ring_push(ctl, slots, /*tag=*/1, /*src=*/blockIdx.x,
              /*arg=*/moved * blockDim.x, t0, t1);
  }
}

static double bench_memcpy(void* dst, const void* src, size_t bytes,
                           hipMemcpyKind kind, int reps) {
  hipStream_t stream;
  HIP_CHECK(hipStreamCreate(&stream));
  // warmup
  HIP_CHECK(hipMemcpyAsync(dst, src, bytes, kind, stream));
  HIP_CHECK(hipStreamSynchronize(stream));
  hipEvent_t e0, e1;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  HIP_CHECK(hipEventRecord(e0, stream));
  for (int r = 0; r < reps; ++r)
    HIP_CHECK(hipMemcpyAsync(dst, src, bytes, kind, stream));
  HIP_CHECK(hipEventRecord(e1, stream));
  HIP_CHECK(hipStreamSynchronize(stream));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
  HIP_CHECK(hipEventDestroy(e0));
  HIP_CHECK(hipEventDestroy(e1));
  HIP_CHECK(hipStreamDestroy(stream));
  return (double) bytes * reps / (ms * 1e-3) / 1e9;  // GB/s
}

int main(int argc, char** argv) {
  int device = 0;
  const char* ring_logdir = nullptr;
  if (argc > 1 && argv[1][0] != '-') device = atoi(argv[1]);
  for (int i = 1; i < argc - 1; ++i)
    if (strcmp(argv[i], "--ring") == 0) ring_logdir = argv[i + 1];
  HIP_CHECK(hipSetDevice(device));
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  printf("# device %d: %s (gcn=%s)\n", device, prop.name, prop.gcnArchName);
  printf("kind,bytes,gbps\n");

  const size_t sizes[] = {1 << 20, 16 << 20, 256 << 20, 1u << 30};
  for (size_t bytes : sizes) {
    int reps = bytes >= (256u << 20) ? 10 : 50;
    void* h_pinned = nullptr;
    void* h_pageable = malloc(bytes);
    void *d_a = nullptr, *d_b = nullptr;
    HIP_CHECK(hipHostMalloc(&h_pinned, bytes));
    HIP_CHECK(hipMalloc(&d_a, bytes));
    HIP_CHECK(hipMalloc(&d_b, bytes));
    memset(h_pageable, 1, bytes);
    memset(h_pinned, 1, bytes);

    printf("H2D_pinned,%zu,%.2f\n", bytes,
           bench_memcpy(d_a, h_pinned, bytes, hipMemcpyHostToDevice, reps));
    printf("H2D_pageable,%zu,%.2f\n", bytes,
           bench_memcpy(d_a, h_pageable, bytes, hipMemcpyHostToDevice, reps));
    printf("D2H_pinned,%zu,%.2f\n", bytes,
           bench_memcpy(h_pinned, d_a, bytes, hipMemcpyDeviceToHost, reps));
    printf("D2D_memcpy,%zu,%.2f\n", bytes,
           bench_memcpy(d_b, d_a, bytes, hipMemcpyDeviceToDevice, reps));

    // kernel copy: read + write = 2x traffic
    {
      size_t n_vec = bytes / sizeof(uint4);
      hipEvent_t e0, e1;
      HIP_CHECK(hipEventCreate(&e0));
      HIP_CHECK(hipEventCreate(&e1));
      dim3 block(256);
      // >> 256 workgroups to fill 256 CUs across 8 XCDs
      dim3 grid(4096);
      hipLaunchKernelGGL(copy_kernel, grid, block, 0, 0, (const uint4*) d_a,
                         (uint4*) d_b, n_vec);
      HIP_CHECK(hipDeviceSynchronize());
      HIP_CHECK(hipEventRecord(e0));
      for (int r = 0; r < reps; ++r)
        hipLaunchKernelGGL(copy_kernel, grid, block, 0, 0, (const uint4*) d_a,
                           (uint4*) d_b, n_vec);
      HIP_CHECK(hipEventRecord(e1));
      HIP_CHECK(hipDeviceSynchronize());
      float ms = 0.f;
      HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
      printf("D2D_kernel_rw,%zu,%.2f\n", bytes,
             2.0 * bytes * reps / (ms * 1e-3) / 1e9);
      HIP_CHECK(hipEventDestroy(e0));
      HIP_CHECK(hipEventDestroy(e1));
    }

    HIP_CHECK(hipHostFree(h_pinned));
    free(h_pageable);
    HIP_CHECK(hipFree(d_a));
    HIP_CHECK(hipFree(d_b));
  }

  if (ring_logdir) {
    // device-trace-ring phase: per-workgroup spans from inside the copy
    // kernel, compacted on-device and landed next to the profiler's records
    const size_t bytes = 256u << 20;
    size_t n_vec = bytes / sizeof(uint4);
    void *d_a = nullptr, *d_b = nullptr;
    HIP_CHECK(hipMalloc(&d_a, bytes));
    HIP_CHECK(hipMalloc(&d_b, bytes));
    HIP_CHECK(hipMemset(d_a, 2, bytes));
    void* ring = nullptr;
    if (sofa_ring_create(device, 1u << 16, &ring) != 0) return 1;
    void *d_ctl = nullptr, *d_slots = nullptr;
    sofa_ring_device_ptrs(ring, &d_ctl, &d_slots);
    dim3 block(256), grid(4096);
    for (int rep = 0; rep < 3; ++rep)
      hipLaunchKernelGGL(copy_kernel_ring, grid, block, 0, 0,
                         (const uint4*) d_a, (uint4*) d_b, n_vec,
                         (RingControl*) d_ctl, (RingRec*) d_slots);
    HIP_CHECK(hipDeviceSynchronize());
    const char* tags[] = {"wg_copy"};
    int n = sofa_ring_dump_sgt(ring, device, ring_logdir, tags, 1);
    printf("devring,%d,records\n", n);
    sofa_ring_destroy(ring);
    HIP_CHECK(hipFree(d_a));
    HIP_CHECK(hipFree(d_b));
  }
  return 0;
}
