// Device-trace ring -> SGT bridge: dump a compacted ring as a gputrace SGT
// file so device-side software events land on the sofa timeline next to the
// rocprofiler-captured kernels.
//
// This is the product path for the on-device trace ring (trace_ring.hip):
// GPU code instrumented with `ring_push` (e.g. per-workgroup phase markers
// inside a persistent kernel — events rocprofiler cannot see) gets its
// records compacted on-device, clock-converted from s_memrealtime ticks to
// CLOCK_MONOTONIC_RAW ns IN the compaction pass, and written as KernelRec
// entries (kernel_id = tag) + KERNEL_NAME/CLOCK records.
//
// Exported C API (ctypes: sofa_amd/record/ring_dump.py):
//   sofa_ring_dump_sgt(ring, logdir, tag_names[], n_tags) -> n_records

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstring>
#include <ctime>
#include <unistd.h>
#include <vector>

#include "../collector/sgt_format.h"

// from trace_ring.hip (same shared object)
extern "C" int sofa_ring_compact(void* ring_p, uint64_t tag_mask, double scale,
                                 long long offset, void* host_out,
                                 uint32_t max_out, uint32_t* n_out);
extern "C" int sofa_ring_head(void* ring_p, unsigned long long* head_out);
extern "C" int sofa_gpu_timebase_sample(int device, uint64_t* host_before_ns,
                                        uint64_t* device_ticks_min,
                                        uint64_t* device_ticks_max,
                                        uint64_t* host_after_ns);
extern "C" int sofa_gpu_timebase_freq(int device, int interval_ms,
                                      double* ticks_per_sec);

namespace {

struct HostRec {  // mirror of trace_ring.hip RingRec
  uint64_t t_start;
  uint64_t t_end;
  uint32_t tag;
  uint32_t src;
  uint64_t arg;
};

uint64_t host_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

void put(std::vector<char>& buf, const void* p, size_t n) {
  const char* c = static_cast<const char*>(p);
  buf.insert(buf.end(), c, c + n);
}

}  // namespace

extern "C" int sofa_ring_dump_sgt(void* ring_p, int device, const char* logdir,
                                  const char* const* tag_names, int n_tags) {
  // clock correlation: device ticks <-> CLOCK_MONOTONIC_RAW via the
  // timebase microkernel (midpoint of the best host window)
  uint64_t hb = 0, dmin = 0, dmax = 0, ha = 0;
  if (sofa_gpu_timebase_sample(device, &hb, &dmin, &dmax, &ha) != 0) return -1;
  double tps = 1e8;
  sofa_gpu_timebase_freq(device, 50, &tps);
  double ns_per_tick = 1e9 / tps;
  double offset = 0.5 * (double) (hb + ha) - (double) dmin * ns_per_tick;

  unsigned long long head = 0;
  sofa_ring_head(ring_p, &head);
  uint32_t cap = head > (1u << 26) ? (1u << 26) : (uint32_t) head;
  std::vector<HostRec> recs(cap ? cap : 1);
  uint32_t n_out = 0;
  // keep all tag classes; convert ticks->mono_raw ns in the pass
  if (sofa_ring_compact(ring_p, ~0ull, ns_per_tick, (long long) offset,
                        recs.data(), cap, &n_out) != 0)
    return -1;
  if (n_out > cap) n_out = cap;

  char path[4096];
  snprintf(path, sizeof(path), "%s/gputrace_ring_%d.sgt", logdir, getpid());
  FILE* f = fopen(path, "wb");
  if (!f) return -1;

  uint64_t rt = host_ns(CLOCK_REALTIME);
  uint64_t mono = host_ns(CLOCK_MONOTONIC_RAW);
  sgt::FileHeader hdr{};
  hdr.magic = sgt::kMagic;
  hdr.version = sgt::kVersion;
  hdr.pid = (uint32_t) getpid();
  hdr.realtime_ns = rt;
  hdr.monotonic_raw_ns = mono;
  hdr.rocp_ns = mono;  // records are in mono_raw ns
  fwrite(&hdr, sizeof(hdr), 1, f);

  std::vector<char> buf;
  sgt::ClockRec cr{};
  cr.h = {sgt::REC_CLOCK, sizeof(sgt::ClockRec), 0};
  cr.realtime_ns = rt;
  cr.monotonic_raw_ns = mono;
  cr.rocp_ns = mono;
  put(buf, &cr, sizeof(cr));

  for (int t = 0; t < n_tags; ++t) {
    const char* nm = tag_names && tag_names[t] ? tag_names[t] : "devring";
    char full[256];
    snprintf(full, sizeof(full), "devring:%s", nm);
    size_t len = strlen(full);
    size_t total = (sizeof(sgt::NameRec) + len + 1 + 7) & ~size_t(7);
    std::vector<char> nb(total, 0);
    auto* rec = reinterpret_cast<sgt::NameRec*>(nb.data());
    rec->h = {sgt::REC_KERNEL_NAME, (uint16_t) total, 0};
    rec->id = 0xD0000000ull + (uint64_t) t;  // devring kernel-id space
    memcpy(nb.data() + sizeof(sgt::NameRec), full, len);
    put(buf, nb.data(), total);
  }

  for (uint32_t i = 0; i < n_out; ++i) {
    sgt::KernelRec kr{};
    kr.h = {sgt::REC_KERNEL, sizeof(sgt::KernelRec), 0};
    kr.start_ns = recs[i].t_start;
    kr.end_ns = recs[i].t_end > recs[i].t_start ? recs[i].t_end : recs[i].t_start + 1;
    kr.corr_id = recs[i].arg;
    kr.tid = recs[i].src;
    kr.device = (uint32_t) device;
    kr.kernel_id = 0xD0000000ull + (recs[i].tag > 0 ? recs[i].tag - 1 : 0) % (n_tags > 0 ? n_tags : 1);
    put(buf, &kr, sizeof(kr));
  }
  fwrite(buf.data(), 1, buf.size(), f);
  fclose(f);
  return (int) n_out;
}
