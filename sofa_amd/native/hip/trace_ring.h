// Shared device-trace-ring types + the ring_push device primitive.
// Producers (any instrumented HIP kernel) and the compaction/dump machinery
// (trace_ring.hip / ring_writer.hip) include this; records are 32-byte
// slots in a device-resident MPMC ring addressed by a monotonic head.
#pragma once

#include <cstdint>

struct RingRec {
  uint64_t t_start;  // device ticks (s_memrealtime) or pre-converted ns
  uint64_t t_end;
  uint32_t tag;      // event class; 0 = empty slot
  uint32_t src;      // producer id (wave/block/stream tag)
  uint64_t arg;      // user payload
};
static_assert(sizeof(RingRec) == 32, "RingRec must be 32 bytes");

struct RingControl {
  unsigned long long head;  // total pushes (monotonic)
  uint32_t capacity;
  uint32_t _pad;
};

#ifdef __HIPCC__
__device__ inline void ring_push(RingControl* ctl, RingRec* slots, uint32_t tag,
                                 uint32_t src, uint64_t arg, uint64_t t0,
                                 uint64_t t1) {
  unsigned long long h = atomicAdd(&ctl->head, 1ull);
  RingRec r;
  r.t_start = t0;
  r.t_end = t1;
  r.tag = tag;
  r.src = src;
  r.arg = arg;
  slots[h % ctl->capacity] = r;
}
#endif
