// sofa-timebase — host clock-correlation helper.
//
// Replaces the reference's bin/sofa_perf_timebase.cc (gettimeofday + a `perf
// record ls` run, :8-19) whose only purpose was pairing the perf uptime clock
// with the epoch clock.  Here every collector stamps CLOCK_MONOTONIC_RAW
// directly (cpusampler via attr.use_clockid; rocprofiler timestamps are
// correlated by the collector library), so this helper just emits repeated
// (REALTIME, MONOTONIC, MONOTONIC_RAW, BOOTTIME) tuples for offset/drift
// estimation in preprocess.
//
// Output: one JSON object per line.

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <ctime>
#include <unistd.h>

static uint64_t ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

int main(int argc, char** argv) {
  int reps = 3;
  if (argc > 1) reps = atoi(argv[1]);
  for (int i = 0; i < reps; i++) {
    uint64_t rt = ns(CLOCK_REALTIME);
    uint64_t mono = ns(CLOCK_MONOTONIC);
    uint64_t raw = ns(CLOCK_MONOTONIC_RAW);
    uint64_t boot = ns(CLOCK_BOOTTIME);
    printf("{\"realtime_ns\": %llu, \"monotonic_ns\": %llu, "
           "\"monotonic_raw_ns\": %llu, \"boottime_ns\": %llu}\n",
           (unsigned long long)rt, (unsigned long long)mono,
           (unsigned long long)raw, (unsigned long long)boot);
    if (i + 1 < reps) usleep(10000);
  }
  return 0;
}
