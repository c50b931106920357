// On-device trace ring + compaction kernel — hand-written HIP for CDNA4.
//
// The reference pipeline collapses at modern kernel-launch rates because
// every GPU event takes a CSV->Python-row round trip (SURVEY.md §7 step 6,
// "event-rate scaling").  The MI355X design keeps a device-resident ring that
// instrumented GPU code (or future device-side tooling) appends 32-byte
// records to with one atomic, and a *compaction kernel* that filters +
// densifies + clock-converts records entirely on device before a single
// coalesced D2H copy (BASELINE.json north star: "on-device trace-ring
// compaction kernel ... LDS staging").
//
// CDNA4 specifics (see /opt/skills/guides/cdna_hip_programming.md):
//  * wavefront = 64: the compaction uses 64-bit __ballot + __popcll
//    prefix-sums (NOT 32-bit warp idioms);
//  * block-level stable compaction: per-wave totals staged through LDS,
//    wave 0 scans, one global atomicAdd per block for the output base;
//  * the tick->ns conversion is fused into the compaction pass (one HBM
//    round trip, not a separate elementwise kernel);
//  * records are 32 B so one lane moves one record as 2x b128-class vector
//    accesses; the output write is dense/coalesced.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define HIP_CHECK(x)                                                          \
  do {                                                                        \
    hipError_t err_ = (x);                                                    \
    if (err_ != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(err_),     \
              __FILE__, __LINE__);                                            \
      return -1;                                                              \
    }                                                                         \
  } while (0)

#include "trace_ring.h"

namespace {

// Test producer: each thread pushes one deterministic record.  Mirrored by
// the CPU reference in tests/test_gpu_trace_ring.py.
__global__ void producer_kernel(RingControl* ctl, RingRec* slots, uint32_t n,
                                uint32_t n_tags) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t t = __builtin_amdgcn_s_memrealtime();
  ring_push(ctl, slots, 1u + (i % n_tags), i / 64, (uint64_t) i * 3u + 1u, t,
            t + 100 + (i % 7));
}

// ------------------------------------------------------- compaction kernel
//
// keep records whose tag-class bit is in tag_mask (bit k = tag class k,
// tag classes are tag % 64); write t' = t * scale + offset (double math —
// scale is ~10.0 for the 100 MHz counter, exact for the < 2^53 tick values
// seen in practice).

constexpr int kBlockThreads = 256;          // 4 waves of 64
constexpr int kWavesPerBlock = kBlockThreads / 64;
constexpr int kRecsPerThread = 8;           // 2048 records per block: the
// original 1-record/thread version spent its time on the 2 __syncthreads +
// 1 global atomic per 256 records (measured 0.9 TB/s); amortizing them over
// 8x the data and scanning per-lane COUNTS (0..8) instead of single-bit
// ballots keeps the output stable in input order.

__global__ void compact_kernel(const RingRec* __restrict__ slots, uint32_t n_valid,
                               uint64_t tag_mask, double scale, long long offset,
                               RingRec* __restrict__ out, uint32_t out_cap,
                               unsigned int* __restrict__ out_count) {
  __shared__ uint32_t wave_totals[kWavesPerBlock];
  __shared__ uint32_t wave_bases[kWavesPerBlock];
  __shared__ unsigned int block_base;

  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  // wave w owns the contiguous span [span, span + 512); within the span,
  // lane l's j-th record is span + j*64 + l — COALESCED loads (for fixed j
  // the 64 lanes read 64 consecutive 32-B records = one contiguous 2 KiB
  // region), unlike thread-consecutive ownership (256-B lane stride).
  // Record order within the wave is (j, lane)-major, so stability falls out
  // of a per-j ballot prefix + a running per-wave total.
  uint32_t span = (blockIdx.x * kWavesPerBlock + wave) * 64 * kRecsPerThread;

  RingRec r[kRecsPerThread];
  bool keep[kRecsPerThread];
#pragma unroll
  for (int j = 0; j < kRecsPerThread; ++j) {
    uint32_t i = span + j * 64 + lane;
    keep[j] = false;
    if (i < n_valid) {
      r[j] = slots[i];
      keep[j] = r[j].tag != 0 && ((tag_mask >> (r[j].tag & 63u)) & 1ull);
    }
  }

  // per-j ballots: output offset of (j, lane) = sum of totals of rounds
  // < j + count of kept lanes < lane in round j
  unsigned long long below = (1ull << lane) - 1ull;
  uint32_t round_base[kRecsPerThread];
  uint32_t lane_prefix[kRecsPerThread];
  uint32_t running = 0;
#pragma unroll
  for (int j = 0; j < kRecsPerThread; ++j) {
    unsigned long long m = __ballot(keep[j]);
    round_base[j] = running;
    lane_prefix[j] = __popcll(m & below);
    running += __popcll(m);
  }
  uint32_t wave_total = running;

  if (lane == 0) wave_totals[wave] = wave_total;
  __syncthreads();

  if (wave == 0 && lane == 0) {
    uint32_t run = 0;
    for (int w = 0; w < kWavesPerBlock; ++w) {
      wave_bases[w] = run;
      run += wave_totals[w];
    }
    block_base = atomicAdd(out_count, run);
  }
  __syncthreads();

  uint32_t base = block_base + wave_bases[wave];
#pragma unroll
  for (int j = 0; j < kRecsPerThread; ++j) {
    if (keep[j]) {
      uint32_t dst = base + round_base[j] + lane_prefix[j];
      if (dst < out_cap) {
        RingRec o = r[j];
        o.t_start = (uint64_t)((double) o.t_start * scale + (double) offset);
        o.t_end = (uint64_t)((double) o.t_end * scale + (double) offset);
        out[dst] = o;
      }
    }
  }
}

struct Ring {
  RingControl* d_ctl;
  RingRec* d_slots;
  uint32_t capacity;
  int device;
};

}  // namespace

extern "C" {

int sofa_ring_create(int device, uint32_t capacity, void** ring_out) {
  HIP_CHECK(hipSetDevice(device));
  Ring* ring = new Ring{};
  ring->capacity = capacity;
  ring->device = device;
  HIP_CHECK(hipMalloc(&ring->d_ctl, sizeof(RingControl)));
  HIP_CHECK(hipMalloc(&ring->d_slots, sizeof(RingRec) * (size_t) capacity));
  RingControl ctl{0, capacity, 0};
  HIP_CHECK(hipMemcpy(ring->d_ctl, &ctl, sizeof(ctl), hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(ring->d_slots, 0, sizeof(RingRec) * (size_t) capacity));
  *ring_out = ring;
  return 0;
}

// Device-side pointers for EXTERNAL producers (instrumented kernels link
// against libsofahip and pass these to their own launches — the product
// path; see bandwidth.hip --ring)
int sofa_ring_device_ptrs(void* ring_p, void** ctl_out, void** slots_out) {
  Ring* ring = static_cast<Ring*>(ring_p);
  *ctl_out = ring->d_ctl;
  *slots_out = ring->d_slots;
  return 0;
}

int sofa_ring_destroy(void* ring_p) {
  Ring* ring = static_cast<Ring*>(ring_p);
  HIP_CHECK(hipFree(ring->d_ctl));
  HIP_CHECK(hipFree(ring->d_slots));
  delete ring;
  return 0;
}

int sofa_ring_test_produce(void* ring_p, uint32_t n, uint32_t n_tags) {
  Ring* ring = static_cast<Ring*>(ring_p);
  HIP_CHECK(hipSetDevice(ring->device));
  dim3 block(256);
  dim3 grid((n + 255) / 256);
  hipLaunchKernelGGL(producer_kernel, grid, block, 0, 0, ring->d_ctl,
                     ring->d_slots, n, n_tags);
  HIP_CHECK(hipDeviceSynchronize());
  return 0;
}

int sofa_ring_head(void* ring_p, unsigned long long* head_out) {
  Ring* ring = static_cast<Ring*>(ring_p);
  RingControl ctl;
  HIP_CHECK(hipMemcpy(&ctl, ring->d_ctl, sizeof(ctl), hipMemcpyDeviceToHost));
  *head_out = ctl.head;
  return 0;
}

// Compact + convert + copy back.  host_out must hold max_out records.
// n_out gets the number of kept records (clamped to max_out on copy).
int sofa_ring_compact(void* ring_p, uint64_t tag_mask, double scale,
                      long long offset, void* host_out, uint32_t max_out,
                      uint32_t* n_out) {
  Ring* ring = static_cast<Ring*>(ring_p);
  HIP_CHECK(hipSetDevice(ring->device));
  RingControl ctl;
  HIP_CHECK(hipMemcpy(&ctl, ring->d_ctl, sizeof(ctl), hipMemcpyDeviceToHost));
  uint32_t n_valid =
      (uint32_t)(ctl.head < ctl.capacity ? ctl.head : ctl.capacity);

  RingRec* d_out = nullptr;
  unsigned int* d_count = nullptr;
  HIP_CHECK(hipMalloc(&d_out, sizeof(RingRec) * (size_t) n_valid));
  HIP_CHECK(hipMalloc(&d_count, sizeof(unsigned int)));
  HIP_CHECK(hipMemset(d_count, 0, sizeof(unsigned int)));

  if (n_valid > 0) {
    dim3 block(kBlockThreads);
    uint32_t per_block = kBlockThreads * kRecsPerThread;
    dim3 grid((n_valid + per_block - 1) / per_block);
    hipLaunchKernelGGL(compact_kernel, grid, block, 0, 0, ring->d_slots,
                       n_valid, tag_mask, scale, offset, d_out, n_valid,
                       d_count);
    HIP_CHECK(hipDeviceSynchronize());
  }

  unsigned int count = 0;
  HIP_CHECK(hipMemcpy(&count, d_count, sizeof(count), hipMemcpyDeviceToHost));
  uint32_t n_copy = count < max_out ? count : max_out;
  if (n_copy > 0) {
    HIP_CHECK(hipMemcpy(host_out, d_out, sizeof(RingRec) * (size_t) n_copy,
                        hipMemcpyDeviceToHost));
  }
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipFree(d_count));
  *n_out = count;
  return 0;
}

// Kernel-only throughput benchmark: run the compaction kernel `iters` times
// (device buffers only, no D2H), return mean kernel ms via hipEvents.
int sofa_ring_compact_bench(void* ring_p, uint64_t tag_mask, int iters,
                            double* ms_out) {
  Ring* ring = static_cast<Ring*>(ring_p);
  HIP_CHECK(hipSetDevice(ring->device));
  RingControl ctl;
  HIP_CHECK(hipMemcpy(&ctl, ring->d_ctl, sizeof(ctl), hipMemcpyDeviceToHost));
  uint32_t n_valid =
      (uint32_t)(ctl.head < ctl.capacity ? ctl.head : ctl.capacity);
  if (n_valid == 0) return -1;
  RingRec* d_out = nullptr;
  unsigned int* d_count = nullptr;
  HIP_CHECK(hipMalloc(&d_out, sizeof(RingRec) * (size_t) n_valid));
  HIP_CHECK(hipMalloc(&d_count, sizeof(unsigned int)));
  dim3 block(kBlockThreads);
  uint32_t per_block = kBlockThreads * kRecsPerThread;
  dim3 grid((n_valid + per_block - 1) / per_block);
  // warmup
  HIP_CHECK(hipMemset(d_count, 0, sizeof(unsigned int)));
  hipLaunchKernelGGL(compact_kernel, grid, block, 0, 0, ring->d_slots, n_valid,
                     tag_mask, 10.0, 12345, d_out, n_valid, d_count);
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  HIP_CHECK(hipEventRecord(e0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(compact_kernel, grid, block, 0, 0, ring->d_slots,
                       n_valid, tag_mask, 10.0, 12345, d_out, n_valid,
                       d_count);
  }
  HIP_CHECK(hipEventRecord(e1));
  HIP_CHECK(hipDeviceSynchronize());
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
  *ms_out = ms / iters;
  HIP_CHECK(hipFree(d_out));
  HIP_CHECK(hipFree(d_count));
  HIP_CHECK(hipEventDestroy(e0));
  HIP_CHECK(hipEventDestroy(e1));
  return 0;
}

}  // extern "C"
