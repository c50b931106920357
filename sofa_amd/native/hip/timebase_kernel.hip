// GPU<->CPU timebase microkernel — hand-written HIP for CDNA4 (gfx950).
//
// Replaces the reference's cuhello.cu trick (cyliustack/sofa bin/cuhello.cu:11-41
// + bin/sofa_preprocess.py:1557-1616: pair the LAST CUPTI activity timestamp
// with the LAST libcuda.so perf sample).  That design has no ROCm analog and
// was redesigned (SURVEY.md §7 "hard parts"): here the device samples its
// wall-clock counter (s_memrealtime, constant-rate 100 MHz on CDNA) directly,
// bracketed by host CLOCK_MONOTONIC_RAW reads, giving a bounded-uncertainty
// (host_before, device, host_after) triple per round.  The minimum-window
// round wins.  A second pass measures the device counter frequency so
// preprocess can convert device ticks -> ns.
//
// The sampling kernel is one wavefront (64 lanes, the CDNA scheduling
// quantum): every lane stamps s_memrealtime into LDS, lane 0 publishes the
// wave-min and wave-max to global memory — the spread is the measurement
// jitter (~tens of ns), recorded so the validator can assert tightness.
// Used by tests/test_gpu_timebase.py and by sofa_record's GPU prologue to
// validate rocprofiler's host-correlated timestamps (BASELINE.json north star).

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <ctime>

#define HIP_CHECK(x)                                                          \
  do {                                                                        \
    hipError_t err_ = (x);                                                    \
    if (err_ != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(err_),     \
              __FILE__, __LINE__);                                            \
      return -1;                                                              \
    }                                                                         \
  } while (0)

namespace {

__global__ void timebase_kernel(uint64_t* out) {
  // one wavefront; per-lane realtime stamps staged through LDS
  __shared__ uint64_t stamps[64];
  int lane = threadIdx.x & 63;
  uint64_t t = __builtin_amdgcn_s_memrealtime();
  stamps[lane] = t;
  __syncthreads();
  if (lane == 0) {
    uint64_t mn = stamps[0], mx = stamps[0];
    for (int i = 1; i < 64; ++i) {
      uint64_t v = stamps[i];
      mn = v < mn ? v : mn;
      mx = v > mx ? v : mx;
    }
    out[0] = mn;
    out[1] = mx;
  }
}

uint64_t host_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

// ---- MFMA-timed marker (BASELINE.json north star: "MFMA-timed markers") --
//
// A marker kernel whose duration is SELF-MEASURED on-device: one wave runs
// `iters` chained v_mfma_f32_16x16x32_bf16 (gfx950 matrix-core op, 4
// independent accumulator chains to keep the MFMA pipe fed) bracketed by
// s_memrealtime stamps.  The tracer's reported span for this kernel can then
// be cross-checked against the kernel's own tick count — a calibrated-
// duration probe that validates collector timestamp scaling end-to-end
// (the reference's cuhello.cu only *created* events; it could not check
// the profiler's clock against ground truth).

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__global__ void mfma_marker_kernel(uint64_t* out, int iters) {
  int lane = threadIdx.x & 63;
  bf16x8 a, b;
  for (int i = 0; i < 8; ++i) {
    a[i] = (__bf16)((lane + i) & 7);
    b[i] = (__bf16)((lane - i) & 7);
  }
  f32x4 c0 = {0, 0, 0, 0}, c1 = {0, 0, 0, 0}, c2 = {0, 0, 0, 0},
        c3 = {0, 0, 0, 0};
  __syncthreads();
  uint64_t t0 = __builtin_amdgcn_s_memrealtime();
  for (int i = 0; i < iters; ++i) {
    c0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c0, 0, 0, 0);
    c1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c1, 0, 0, 0);
    c2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c2, 0, 0, 0);
    c3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c3, 0, 0, 0);
  }
  uint64_t t1 = __builtin_amdgcn_s_memrealtime();
  if (lane == 0) {
    out[0] = t0;
    out[1] = t1;
  }
  // keep the accumulators live
  float s = c0[0] + c1[1] + c2[2] + c3[3];
  if (s == 1234567.0f) out[2] = 1;  // never taken; defeats DCE
}

}  // namespace

extern "C" {

// One correlation round: returns 0 on success and fills
//   host_before_ns / host_after_ns  (CLOCK_MONOTONIC_RAW)
//   device_ticks_min / device_ticks_max (s_memrealtime, raw ticks)
// The kernel is pre-warmed by the caller (first launch pays setup).
int sofa_gpu_timebase_sample(int device, uint64_t* host_before_ns,
                             uint64_t* device_ticks_min,
                             uint64_t* device_ticks_max,
                             uint64_t* host_after_ns) {
  HIP_CHECK(hipSetDevice(device));
  uint64_t* d_out = nullptr;
  HIP_CHECK(hipMalloc(&d_out, 2 * sizeof(uint64_t)));
  // warm up: hide one-time launch setup outside the timed window
  hipLaunchKernelGGL(timebase_kernel, dim3(1), dim3(64), 0, 0, d_out);
  HIP_CHECK(hipDeviceSynchronize());

  uint64_t before = host_ns(CLOCK_MONOTONIC_RAW);
  hipLaunchKernelGGL(timebase_kernel, dim3(1), dim3(64), 0, 0, d_out);
  HIP_CHECK(hipDeviceSynchronize());
  uint64_t after = host_ns(CLOCK_MONOTONIC_RAW);

  uint64_t h_out[2] = {0, 0};
  HIP_CHECK(hipMemcpy(h_out, d_out, sizeof(h_out), hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(d_out));

  *host_before_ns = before;
  *host_after_ns = after;
  *device_ticks_min = h_out[0];
  *device_ticks_max = h_out[1];
  return 0;
}

// Measure the s_memrealtime tick rate against CLOCK_MONOTONIC_RAW over
// interval_ms.  Returns ticks per second (expected ~1e8 on CDNA).
int sofa_gpu_timebase_freq(int device, int interval_ms, double* ticks_per_sec) {
  uint64_t hb0, dmin0, dmax0, ha0;
  uint64_t hb1, dmin1, dmax1, ha1;
  if (sofa_gpu_timebase_sample(device, &hb0, &dmin0, &dmax0, &ha0) != 0)
    return -1;
  struct timespec ts = {interval_ms / 1000, (interval_ms % 1000) * 1000000L};
  nanosleep(&ts, nullptr);
  if (sofa_gpu_timebase_sample(device, &hb1, &dmin1, &dmax1, &ha1) != 0)
    return -1;
  double host_dt = 0.5 * ((hb1 + ha1) - (hb0 + ha0));  // midpoint difference, ns
  double dev_dt = double(dmin1 - dmin0);
  if (host_dt <= 0) return -1;
  *ticks_per_sec = dev_dt / (host_dt * 1e-9);
  return 0;
}

// MFMA-timed marker: launches the self-timed matrix-core burst; fills the
// device tick pair + the host MONOTONIC_RAW window around the launch, and
// the implied on-device duration in ns (ticks * 10 at the CDNA 100 MHz
// constant counter).  A tracer profiling this process will report a span for
// `mfma_marker_kernel` that must agree with self_ns — the validation is done
// by analyze (clock_validation feature).
int sofa_gpu_mfma_marker(int device, int iters, uint64_t* host_before_ns,
                         uint64_t* host_after_ns, uint64_t* self_ticks,
                         double* self_ns) {
  HIP_CHECK(hipSetDevice(device));
  uint64_t* d_out = nullptr;
  HIP_CHECK(hipMalloc(&d_out, 4 * sizeof(uint64_t)));
  hipLaunchKernelGGL(mfma_marker_kernel, dim3(1), dim3(64), 0, 0, d_out, 16);
  HIP_CHECK(hipDeviceSynchronize());  // warm: code-object load + MFMA clock-up

  uint64_t before = host_ns(CLOCK_MONOTONIC_RAW);
  hipLaunchKernelGGL(mfma_marker_kernel, dim3(1), dim3(64), 0, 0, d_out, iters);
  HIP_CHECK(hipDeviceSynchronize());
  uint64_t after = host_ns(CLOCK_MONOTONIC_RAW);

  uint64_t h_out[2] = {0, 0};
  HIP_CHECK(hipMemcpy(h_out, d_out, sizeof(h_out), hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(d_out));
  *host_before_ns = before;
  *host_after_ns = after;
  *self_ticks = h_out[1] - h_out[0];
  *self_ns = double(h_out[1] - h_out[0]) * 10.0;  // 100 MHz constant counter
  return 0;
}

}  // extern "C"
