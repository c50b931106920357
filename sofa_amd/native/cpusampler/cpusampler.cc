// sofa-cpusampler — native CPU sampling profiler on perf_event_open(2).
//
// Replaces the reference's dependency on the external `perf record -F 99`
// binary + `perf script` text pipeline (cyliustack/sofa bin/sofa_record.py:339-354,
// bin/sofa_preprocess.py:405-414,1791-1799) with a self-contained sampler that
// writes a compact binary stream parsed vectorized in preprocess.
//
// Design notes (MI355X/Linux-native, not a port):
//  * samples the software cpu-clock event (VMs here expose no hardware PMU),
//    frequency-mode at -F Hz (default 99, the reference's rate);
//  * perf timestamps are forced onto CLOCK_MONOTONIC_RAW via
//    attr.use_clockid — this removes the reference's entire
//    "perf-uptime <-> unixtime" two-binary clock pairing dance
//    (bin/sofa_perf_timebase.cc, bin/sofa_preprocess.py:1553-1616):
//    the output header carries one (REALTIME, MONOTONIC_RAW) pair.
//  * PERF_RECORD_MMAP2 + COMM + TASK records are captured so preprocess can
//    symbolize IPs offline (per-pid address-space reconstruction).
//
// Output format "SCS1": header, then length-prefixed records (see structs).
//
// Usage: sofa-cpusampler -o out.scs [-F 99] (-a | -p PID) [--max-mb N]
//   Runs until SIGTERM/SIGINT (or parent-death); flushes and exits.

#include <linux/perf_event.h>
#include <sys/ioctl.h>
#include <sys/mman.h>
#include <sys/prctl.h>
#include <sys/syscall.h>
#include <poll.h>
#include <signal.h>
#include <unistd.h>
#include <fcntl.h>

#include <cerrno>
#include <cinttypes>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <string>
#include <vector>

namespace {

constexpr uint32_t kMagic = 0x31534353;  // "SCS1"

struct FileHeader {
  uint32_t magic;
  uint32_t version;
  uint64_t realtime_ns;       // CLOCK_REALTIME at start
  uint64_t monotonic_raw_ns;  // CLOCK_MONOTONIC_RAW at start (sample clock)
  uint64_t boottime_ns;       // CLOCK_BOOTTIME at start
  uint32_t sample_freq;
  uint32_t n_cpus;
  uint64_t reserved[4];
};

enum RecType : uint16_t {
  REC_SAMPLE = 1,
  REC_MMAP = 2,
  REC_COMM = 3,
  REC_EXIT = 4,
  REC_LOST = 5,
  REC_SAMPLE_CS = 6,  // sample + fixed-depth callchain (-g mode)
};

constexpr int kMaxFrames = 16;

struct RecHeader {
  uint16_t type;
  uint16_t size;  // total bytes including header
};

struct SampleRec {  // REC_SAMPLE
  RecHeader h;
  uint64_t time_ns;
  uint64_t ip;
  uint32_t pid, tid;
  uint32_t cpu;
  uint32_t flags;  // bit0: kernel-space sample
  uint64_t period;
};

struct SampleCsRec {  // REC_SAMPLE_CS: fixed size for vectorized parsing
  RecHeader h;
  uint64_t time_ns;
  uint64_t ip;
  uint32_t pid, tid;
  uint32_t cpu;
  uint32_t flags;
  uint64_t period;
  uint32_t n_frames;
  uint32_t _pad;
  uint64_t frames[kMaxFrames];  // user-space return addresses, leaf first
};

struct MmapRec {  // REC_MMAP, followed by filename (size-derived length)
  RecHeader h;
  uint64_t time_ns;
  uint32_t pid, tid;
  uint64_t addr, len, pgoff;
  // char filename[];
};

struct CommRec {  // REC_COMM, followed by comm string
  RecHeader h;
  uint64_t time_ns;
  uint32_t pid, tid;
  // char comm[];
};

struct LostRec {  // REC_LOST
  RecHeader h;
  uint64_t time_ns;
  uint64_t lost;
};

volatile sig_atomic_t g_stop = 0;
void on_signal(int) { g_stop = 1; }

long perf_event_open(struct perf_event_attr* a, pid_t pid, int cpu, int gfd,
                     unsigned long flags) {
  return syscall(SYS_perf_event_open, a, pid, cpu, gfd, flags);
}

uint64_t clock_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

class Writer {
 public:
  explicit Writer(FILE* f, uint64_t max_bytes)
      : f_(f), max_bytes_(max_bytes) {
    buf_.reserve(1 << 20);
  }
  void append(const void* p, size_t n) {
    if (written_ + buf_.size() + n > max_bytes_) return;  // cap output
    const char* c = static_cast<const char*>(p);
    buf_.insert(buf_.end(), c, c + n);
    if (buf_.size() >= (1 << 20)) flush();
  }
  void flush() {
    if (!buf_.empty()) {
      fwrite(buf_.data(), 1, buf_.size(), f_);
      written_ += buf_.size();
      buf_.clear();
    }
    fflush(f_);
  }

 private:
  FILE* f_;
  uint64_t max_bytes_;
  uint64_t written_ = 0;
  std::vector<char> buf_;
};

struct Ring {
  int fd = -1;
  struct perf_event_mmap_page* meta = nullptr;
  char* data = nullptr;
  size_t data_size = 0;
};

// perf sample layout given our sample_type =
// IP | TID | TIME | CPU | PERIOD (in that order per ABI); with -g,
// CALLCHAIN (u64 nr; u64 ips[nr]) follows PERIOD.
struct RawSample {
  uint64_t ip;
  uint32_t pid, tid;
  uint64_t time;
  uint32_t cpu, res;
  uint64_t period;
};

bool g_callchain = false;

void drain_ring(Ring& r, Writer& w) {
  auto* m = r.meta;
  uint64_t head = __atomic_load_n(&m->data_head, __ATOMIC_ACQUIRE);
  uint64_t tail = m->data_tail;
  const size_t mask = r.data_size - 1;
  std::vector<char> tmp;
  while (tail < head) {
    // the 8-byte record header itself may wrap the ring boundary
    struct perf_event_header hdr_copy;
    const struct perf_event_header* eh;
    size_t hpos = tail & mask;
    if (hpos + sizeof(hdr_copy) > r.data_size) {
      size_t first = r.data_size - hpos;
      memcpy(&hdr_copy, r.data + hpos, first);
      memcpy(reinterpret_cast<char*>(&hdr_copy) + first, r.data,
             sizeof(hdr_copy) - first);
      eh = &hdr_copy;
    } else {
      eh = reinterpret_cast<struct perf_event_header*>(r.data + hpos);
    }
    uint16_t esz = eh->size;
    if (esz == 0) break;  // corrupt ring: bail rather than spin
    const char* ev;
    if (((tail & mask) + esz) > r.data_size) {
      // wrapped event: copy into tmp
      tmp.resize(esz);
      size_t first = r.data_size - (tail & mask);
      memcpy(tmp.data(), r.data + (tail & mask), first);
      memcpy(tmp.data() + first, r.data, esz - first);
      ev = tmp.data();
    } else {
      ev = r.data + (tail & mask);
    }
    auto* ph = reinterpret_cast<const struct perf_event_header*>(ev);
    const char* body = ev + sizeof(struct perf_event_header);
    switch (ph->type) {
      case PERF_RECORD_SAMPLE: {
        const auto* s = reinterpret_cast<const RawSample*>(body);
        uint32_t kflag = (ph->misc & PERF_RECORD_MISC_CPUMODE_MASK) ==
                                 PERF_RECORD_MISC_KERNEL
                             ? 1u
                             : 0u;
        if (!g_callchain) {
          SampleRec rec{};
          rec.h = {REC_SAMPLE, sizeof(SampleRec)};
          rec.time_ns = s->time;
          rec.ip = s->ip;
          rec.pid = s->pid;
          rec.tid = s->tid;
          rec.cpu = s->cpu;
          rec.flags = kflag;
          rec.period = s->period;
          w.append(&rec, sizeof(rec));
        } else {
          SampleCsRec rec{};
          rec.h = {REC_SAMPLE_CS, sizeof(SampleCsRec)};
          rec.time_ns = s->time;
          rec.ip = s->ip;
          rec.pid = s->pid;
          rec.tid = s->tid;
          rec.cpu = s->cpu;
          rec.flags = kflag;
          rec.period = s->period;
          const char* cc = body + sizeof(RawSample);
          uint64_t nr = 0;
          memcpy(&nr, cc, sizeof(nr));
          const uint64_t* ips = reinterpret_cast<const uint64_t*>(cc + 8);
          uint32_t out_n = 0;
          for (uint64_t f = 0; f < nr && out_n < kMaxFrames; ++f) {
            // skip PERF_CONTEXT_* markers ((u64)-128 .. (u64)-4096 sentinels)
            if (ips[f] >= 0xfffffffffffff000ull) continue;
            rec.frames[out_n++] = ips[f];
          }
          rec.n_frames = out_n;
          w.append(&rec, sizeof(rec));
        }
        break;
      }
      case PERF_RECORD_MMAP2: {
        // body: pid,tid,addr,len,pgoff, (maj,min,ino,ino_gen)|(build-id),
        // prot,flags, filename..., then sample_id trailer
        struct M2 {
          uint32_t pid, tid;
          uint64_t addr, len, pgoff;
          uint8_t misc[24];
          uint32_t prot, flags2;
        };
        const auto* mm = reinterpret_cast<const M2*>(body);
        const char* fname = body + sizeof(M2);
        size_t fmax = esz - sizeof(struct perf_event_header) - sizeof(M2);
        size_t flen = strnlen(fname, fmax);
        // executable mappings only (reduce noise): prot & PROT_EXEC
        if (!(mm->prot & 4)) break;
        size_t total = (sizeof(MmapRec) + flen + 1 + 7) & ~size_t(7);
        std::vector<char> out(total, 0);
        auto* rec = reinterpret_cast<MmapRec*>(out.data());
        rec->h = {REC_MMAP, static_cast<uint16_t>(total)};
        rec->time_ns = 0;  // mmap records carry no time with our sample_id_all=0
        rec->pid = mm->pid;
        rec->tid = mm->tid;
        rec->addr = mm->addr;
        rec->len = mm->len;
        rec->pgoff = mm->pgoff;
        memcpy(out.data() + sizeof(MmapRec), fname, flen);
        out[sizeof(MmapRec) + flen] = 0;
        w.append(out.data(), total);
        break;
      }
      case PERF_RECORD_COMM: {
        struct C {
          uint32_t pid, tid;
        };
        const auto* cm = reinterpret_cast<const C*>(body);
        const char* comm = body + sizeof(C);
        size_t cmax = esz - sizeof(struct perf_event_header) - sizeof(C);
        size_t clen = strnlen(comm, cmax);
        size_t total = (sizeof(CommRec) + clen + 1 + 7) & ~size_t(7);
        std::vector<char> out(total, 0);
        auto* rec = reinterpret_cast<CommRec*>(out.data());
        rec->h = {REC_COMM, static_cast<uint16_t>(total)};
        rec->time_ns = 0;
        rec->pid = cm->pid;
        rec->tid = cm->tid;
        memcpy(out.data() + sizeof(CommRec), comm, clen);
        out[sizeof(CommRec) + clen] = 0;
        w.append(out.data(), total);
        break;
      }
      case PERF_RECORD_LOST: {
        struct L {
          uint64_t id, lost;
        };
        const auto* ll = reinterpret_cast<const L*>(body);
        LostRec rec{};
        rec.h = {REC_LOST, sizeof(LostRec)};
        rec.time_ns = 0;
        rec.lost = ll->lost;
        w.append(&rec, sizeof(rec));
        break;
      }
      default:
        break;
    }
    tail += esz;
  }
  __atomic_store_n(&m->data_tail, tail, __ATOMIC_RELEASE);
}

}  // namespace

int main(int argc, char** argv) {
  const char* out_path = nullptr;
  int freq = 99;
  pid_t target_pid = -1;  // -1 => system-wide
  bool system_wide = false;
  const char* cgroup_path = nullptr;  // -G: scope to one cgroup (container)
  bool want_hw_cycles = false;  // -e cycles: try the PMU, fall back to sw
  uint64_t max_mb = 512;
  for (int i = 1; i < argc; i++) {
    std::string a = argv[i];
    if (a == "-o" && i + 1 < argc) out_path = argv[++i];
    else if (a == "-F" && i + 1 < argc) freq = atoi(argv[++i]);
    else if (a == "-p" && i + 1 < argc) target_pid = atoi(argv[++i]);
    else if (a == "-a") system_wide = true;
    else if (a == "-G" && i + 1 < argc) cgroup_path = argv[++i];
    else if (a == "--max-mb" && i + 1 < argc) max_mb = strtoull(argv[++i], nullptr, 10);
    else if (a == "-g" || a == "--callchain") g_callchain = true;
    else if (a == "-e" && i + 1 < argc) {
      want_hw_cycles = (std::string(argv[++i]) == "cycles");
    }
    else {
      fprintf(stderr, "usage: %s -o out.scs [-F hz] (-a | -p pid) [--max-mb N]\n", argv[0]);
      return 2;
    }
  }
  if (!out_path || (!system_wide && target_pid < 0 && !cgroup_path)) {
    fprintf(stderr, "sofa-cpusampler: need -o and one of -a / -p / -G\n");
    return 2;
  }
  // -G <cgroup dir>: profile only tasks in that cgroup (container target,
  // reference `perf record --cgroup=docker/<cid>`, bin/sofa_record.py:394)
  int cgroup_fd = -1;
  if (cgroup_path) {
    cgroup_fd = open(cgroup_path, O_RDONLY | O_DIRECTORY | O_CLOEXEC);
    if (cgroup_fd < 0) {
      fprintf(stderr, "sofa-cpusampler: open cgroup %s: %s\n", cgroup_path,
              strerror(errno));
      return 2;
    }
  }

  signal(SIGTERM, on_signal);
  signal(SIGINT, on_signal);
  prctl(PR_SET_PDEATHSIG, SIGTERM);  // die with the recorder

  int n_cpus = static_cast<int>(sysconf(_SC_NPROCESSORS_ONLN));
  FILE* f = fopen(out_path, "wb");
  if (!f) {
    perror("sofa-cpusampler: fopen");
    return 1;
  }

  struct perf_event_attr attr;
  memset(&attr, 0, sizeof(attr));
  attr.type = PERF_TYPE_SOFTWARE;
  attr.size = sizeof(attr);
  attr.config = PERF_COUNT_SW_CPU_CLOCK;
  if (want_hw_cycles) {
    // probe the PMU once; VMs commonly expose none -> keep sw cpu-clock
    struct perf_event_attr probe;
    memset(&probe, 0, sizeof(probe));
    probe.type = PERF_TYPE_HARDWARE;
    probe.size = sizeof(probe);
    probe.config = PERF_COUNT_HW_CPU_CYCLES;
    probe.disabled = 1;
    int pfd = (int) perf_event_open(&probe, 0, -1, -1, PERF_FLAG_FD_CLOEXEC);
    if (pfd >= 0) {
      close(pfd);
      attr.type = PERF_TYPE_HARDWARE;
      attr.config = PERF_COUNT_HW_CPU_CYCLES;
    } else {
      fprintf(stderr, "sofa-cpusampler: no hardware PMU; using cpu-clock\n");
    }
  }

  FileHeader hdr{};
  hdr.magic = kMagic;
  hdr.version = 1;
  hdr.realtime_ns = clock_ns(CLOCK_REALTIME);
  hdr.monotonic_raw_ns = clock_ns(CLOCK_MONOTONIC_RAW);
  hdr.boottime_ns = clock_ns(CLOCK_BOOTTIME);
  hdr.sample_freq = static_cast<uint32_t>(freq);
  hdr.n_cpus = static_cast<uint32_t>(n_cpus);
  // reserved[0] = sampled event unit so preprocess can convert periods:
  // 0 = sw cpu-clock (period in ns), 1 = hw cycles (period in cycles;
  // divide by MHz from cpuinfo.txt).  Round-1 ADVICE: without this, cycles
  // periods were read as ns and inflated CPU busy time by the clock freq.
  hdr.reserved[0] = (attr.type == PERF_TYPE_HARDWARE) ? 1 : 0;
  fwrite(&hdr, sizeof(hdr), 1, f);

  Writer writer(f, max_mb * (1ull << 20));
  attr.freq = 1;
  attr.sample_freq = static_cast<uint64_t>(freq);
  attr.sample_type =
      PERF_SAMPLE_IP | PERF_SAMPLE_TID | PERF_SAMPLE_TIME | PERF_SAMPLE_CPU |
      PERF_SAMPLE_PERIOD;
  attr.disabled = 1;
  attr.inherit = (target_pid > 0) ? 1 : 0;
  attr.mmap = 1;
  attr.mmap2 = 1;
  attr.comm = 1;
  attr.task = 1;
  attr.exclude_kernel = 0;
  attr.use_clockid = 1;
  attr.clockid = CLOCK_MONOTONIC_RAW;
  attr.wakeup_events = 16;
  if (g_callchain) {
    attr.sample_type |= PERF_SAMPLE_CALLCHAIN;
    attr.sample_max_stack = 32;
  }

  std::vector<Ring> rings;
  const size_t n_pages = 64;  // 64 data pages = 256 KiB per CPU
  const size_t page = static_cast<size_t>(sysconf(_SC_PAGESIZE));
  for (int cpu = 0; cpu < n_cpus; cpu++) {
    Ring r;
    unsigned long flags = PERF_FLAG_FD_CLOEXEC;
    pid_t open_pid = system_wide ? -1 : target_pid;
    if (cgroup_fd >= 0) {
      open_pid = cgroup_fd;  // perf cgroup mode: pid = cgroup dir fd, per-cpu
      flags |= PERF_FLAG_PID_CGROUP;
    }
    r.fd = static_cast<int>(perf_event_open(&attr, open_pid, cpu, -1, flags));
    if (r.fd < 0) {
      if (errno == ENODEV) continue;  // offline cpu
      fprintf(stderr, "sofa-cpusampler: perf_event_open cpu%d: %s\n", cpu,
              strerror(errno));
      continue;
    }
    void* m = mmap(nullptr, (n_pages + 1) * page, PROT_READ | PROT_WRITE,
                   MAP_SHARED, r.fd, 0);
    if (m == MAP_FAILED) {
      fprintf(stderr, "sofa-cpusampler: mmap cpu%d: %s\n", cpu, strerror(errno));
      close(r.fd);
      continue;
    }
    r.meta = static_cast<struct perf_event_mmap_page*>(m);
    r.data = static_cast<char*>(m) + page;
    r.data_size = n_pages * page;
    rings.push_back(r);
  }
  if (rings.empty()) {
    fprintf(stderr, "sofa-cpusampler: no perf events opened\n");
    fclose(f);
    return 1;
  }

  // Snapshot pre-existing executable mappings of the target (perf only emits
  // MMAP2 for *new* mappings after enable).
  auto snapshot_maps = [&](pid_t pid) {
    char path[64];
    snprintf(path, sizeof(path), "/proc/%d/maps", pid);
    FILE* mf = fopen(path, "r");
    if (!mf) return;
    char line[4096];
    while (fgets(line, sizeof(line), mf)) {
      uint64_t lo, hi, off;
      char perms[8];
      char fname[3584];
      fname[0] = 0;
      int n = sscanf(line, "%" SCNx64 "-%" SCNx64 " %7s %" SCNx64 " %*s %*s %3583[^\n]",
                     &lo, &hi, perms, &off, fname);
      if (n < 4 || strchr(perms, 'x') == nullptr || fname[0] != '/') continue;
      size_t flen = strlen(fname);
      size_t total = (sizeof(MmapRec) + flen + 1 + 7) & ~size_t(7);
      std::vector<char> out(total, 0);
      auto* rec = reinterpret_cast<MmapRec*>(out.data());
      rec->h = {REC_MMAP, static_cast<uint16_t>(total)};
      rec->time_ns = 0;
      rec->pid = static_cast<uint32_t>(pid);
      rec->tid = static_cast<uint32_t>(pid);
      rec->addr = lo;
      rec->len = hi - lo;
      rec->pgoff = off;
      memcpy(out.data() + sizeof(MmapRec), fname, flen + 1);
      writer.append(out.data(), total);
    }
    fclose(mf);
  };
  if (target_pid > 0) snapshot_maps(target_pid);

  for (auto& r : rings) ioctl(r.fd, PERF_EVENT_IOC_ENABLE, 0);

  std::vector<struct pollfd> pfds;
  for (auto& r : rings) pfds.push_back({r.fd, POLLIN, 0});

  while (!g_stop) {
    int rc = poll(pfds.data(), pfds.size(), 200);
    if (rc < 0 && errno != EINTR) break;
    for (auto& r : rings) drain_ring(r, writer);
    // exit when target is gone
    if (target_pid > 0 && kill(target_pid, 0) != 0) break;
  }
  for (auto& r : rings) {
    ioctl(r.fd, PERF_EVENT_IOC_DISABLE, 0);
    drain_ring(r, writer);
  }
  writer.flush();
  fclose(f);
  return 0;
}
