// libsofahsalite — low-overhead HSA-level GPU kernel/copy tracer.
//
// WHY (round-1 verdict): the rocprofiler-sdk collector's measured overhead on
// ResNet-50 bs=64 is ~13% with a ~11% floor attributable to the SDK's
// dispatch interception itself (null-sink study, profiles/overhead_final_r01).
// This library goes below the SDK: it is a ROCr tools library (HSA_TOOLS_LIB)
// that wraps the HSA API table directly — the same mechanism rocprofiler sits
// on, minus its per-dispatch machinery.  Design for minimal critical-path
// cost per kernel launch:
//
//   * AQL packets are observed via hsa_amd_queue_intercept_create; the
//     submit-side handler only (a) pops a pre-created signal from a per-queue
//     pool, (b) writes it into the packet's completion_signal if the packet
//     has none, (c) stashes a 64-byte pending slot.  No allocation, no
//     syscall, no interrupt setup per dispatch.
//   * Timing signals are HSA_AMD_SIGNAL_AMD_GPU_ONLY: the packet processor
//     stamps start/end ticks into the signal and decrements it WITHOUT
//     raising a host interrupt — the round-1 suspicion is that per-dispatch
//     interrupt + callback-thread wakeups are the SDK's floor.
//   * A reaper thread polls pending slots (plain memory reads) off the
//     critical path, converts ticks -> ns, and writes SGT KernelRec records
//     (sgt_format.h) identical to the SDK collector's, so preprocess is
//     unchanged.
//   * Packets that already carry a completion signal (HIP sync points — the
//     rare case) are left untouched by default: no semantic risk, host-side
//     bookkeeping only.  SOFA_LITE_REPLACE_SIGNALS=1 opts into replacing
//     those too (our signal in the packet, original decremented at reap).
//   * Async SDMA copies are traced by wrapping hsa_amd_memory_async_copy
//     (+_on_engine/_rect): our interrupt signal goes to the real call, an
//     async handler reads hsa_amd_profiling_get_async_copy_time and forwards
//     the caller's completion signal immediately (copies are low-rate; the
//     ~10 us handler latency is irrelevant next to >100 us SDMA transfers).
//
// Kernel names come from wrapping hsa_executable_freeze and iterating kernel
// symbols: NameRec(kernel_object -> mangled name); KernelRec.kernel_id is the
// kernel_object handle.
//
// Overhead decomposition knobs (measured on MI355X, see profiles/):
//   SOFA_LITE_MODE=off      intercept queues, forward untouched (proxy floor)
//   SOFA_LITE_MODE=prof     + hsa_amd_profiling_set_profiler_enabled only
//   SOFA_LITE_MODE=full     + signal attach/reap = the product (default)
//   SOFA_DEFER_START=1      arm via sofa_lite_start() (bench A/B)
//
// Reference lineage: replaces the system-wide nvprof/CUPTI capture of
// cyliustack/sofa (bin/sofa_record.py:217-242); the SDK collector
// (collector/sofatracer.cc) remains the full-fidelity mode (HIP API spans,
// RCCL args, KFD events) and the two compose: SOFA_TRACE_DISPATCH=0 turns
// the SDK collector into an RCCL/marker-only tracer with no dispatch cost.

#include <hsa/hsa.h>
#include <hsa/hsa_api_trace.h>
#include <hsa/hsa_ext_amd.h>

#include <pthread.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "../collector/sgt_format.h"

namespace {

// ------------------------------------------------------------------ plumbing

CoreApiTable g_core{};     // saved original function pointers
AmdExtTable g_amd{};
bool g_saved = false;

FILE* g_out = nullptr;
std::mutex g_out_mutex;
std::atomic<uint64_t> g_n_records{0};
std::atomic<bool> g_armed{true};
std::atomic<bool> g_shutdown{false};
// set once the REAL hsa_shut_down has run: our cached table pointers skip
// ROCr's public liveness guards, so every wrapper must refuse calls on a
// dead runtime (ROCclr's Device::~Device destroys queues after shutdown —
// measured: that tail call segfaulted)
std::atomic<bool> g_runtime_down{false};

enum LiteMode { MODE_OFF = 0, MODE_PROF = 1, MODE_FULL = 2 };
LiteMode g_mode = MODE_FULL;
bool g_replace_signals = false;

double g_tick_to_ns = 10.0;  // 1e9 / HSA_SYSTEM_INFO_TIMESTAMP_FREQUENCY

uint64_t host_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

uint32_t my_tid() {
  static thread_local uint32_t tid = (uint32_t) syscall(SYS_gettid);
  return tid;
}

bool env_flag(const char* name, bool dflt) {
  const char* v = getenv(name);
  if (!v || !*v) return dflt;
  return !(v[0] == '0' || v[0] == 'n' || v[0] == 'N' || v[0] == 'f' || v[0] == 'F');
}

uint64_t sys_now_ns() {
  uint64_t t = 0;
  g_core.hsa_system_get_info_fn(HSA_SYSTEM_INFO_TIMESTAMP, &t);
  return (uint64_t) (t * g_tick_to_ns);
}

void write_raw(const void* p, size_t n) {
  std::lock_guard<std::mutex> lk(g_out_mutex);
  if (g_out) fwrite(p, 1, n, g_out);
}

constexpr size_t kMaxNameLen = 65000;

void write_name_rec(uint16_t type, uint64_t id, const char* name, size_t len) {
  if (len > kMaxNameLen) len = kMaxNameLen;
  size_t total = (sizeof(sgt::NameRec) + len + 1 + 7) & ~size_t(7);
  std::vector<char> buf(total, 0);
  auto* rec = reinterpret_cast<sgt::NameRec*>(buf.data());
  rec->h = {type, static_cast<uint16_t>(total), 0};
  rec->id = id;
  if (len) memcpy(buf.data() + sizeof(sgt::NameRec), name, len);
  write_raw(buf.data(), total);
}

void write_opname_rec(uint32_t kind, uint32_t op, const char* name) {
  size_t len = strlen(name);
  size_t total = (sizeof(sgt::OpNameRec) + len + 1 + 7) & ~size_t(7);
  std::vector<char> buf(total, 0);
  auto* rec = reinterpret_cast<sgt::OpNameRec*>(buf.data());
  rec->h = {sgt::REC_OPNAME, static_cast<uint16_t>(total), 0};
  rec->kind = kind;
  rec->op = op;
  memcpy(buf.data() + sizeof(sgt::OpNameRec), name, len);
  write_raw(buf.data(), total);
}

void write_clock_rec() {
  sgt::ClockRec rec{};
  rec.h = {sgt::REC_CLOCK, sizeof(sgt::ClockRec), 0};
  rec.realtime_ns = host_ns(CLOCK_REALTIME);
  rec.monotonic_raw_ns = host_ns(CLOCK_MONOTONIC_RAW);
  rec.rocp_ns = sys_now_ns();
  write_raw(&rec, sizeof(rec));
}

// ------------------------------------------------------------------- agents

struct AgentInfo {
  int32_t device = -1;  // GPU index in HSA enumeration order; -1 for CPU
  hsa_agent_t agent{};
};
// LEAKED on purpose: this library's static destructors run before
// ROCclr's RuntimeTearDown (we are dlopened after libamdhip64), and
// ROCclr still calls our queue wrappers from its teardown — iterating a
// destructed container was the round-2 exit crash (gdb: q=0x555000e3bb91).
std::unordered_map<uint64_t, AgentInfo>& agents_map() {
  static auto* m = new std::unordered_map<uint64_t, AgentInfo>();
  return *m;
}
std::mutex g_agents_mutex;
std::atomic<bool> g_agents_done{false};

void enumerate_agents() {
  std::lock_guard<std::mutex> lk(g_agents_mutex);
  if (g_agents_done.load()) return;
  struct Ctx {
    int gpu_idx = 0;
  } ctx;
  g_core.hsa_iterate_agents_fn(
      [](hsa_agent_t agent, void* vctx) -> hsa_status_t {
        auto* c = static_cast<Ctx*>(vctx);
        hsa_device_type_t type = HSA_DEVICE_TYPE_CPU;
        g_core.hsa_agent_get_info_fn(agent, HSA_AGENT_INFO_DEVICE, &type);
        AgentInfo info;
        info.agent = agent;
        if (type == HSA_DEVICE_TYPE_GPU) {
          info.device = c->gpu_idx++;
          sgt::AgentRec rec{};
          rec.h = {sgt::REC_AGENT, sizeof(sgt::AgentRec), 0};
          rec.agent_handle = agent.handle;
          rec.device = info.device;
          rec.type = 2;  // rocprofiler_agent_type GPU
          uint32_t wfs = 64, cus = 0, xcc = 1, node = 0;
          g_core.hsa_agent_get_info_fn(agent, HSA_AGENT_INFO_WAVEFRONT_SIZE, &wfs);
          g_core.hsa_agent_get_info_fn(
              agent, (hsa_agent_info_t) HSA_AMD_AGENT_INFO_COMPUTE_UNIT_COUNT, &cus);
          g_core.hsa_agent_get_info_fn(
              agent, (hsa_agent_info_t) HSA_AMD_AGENT_INFO_NUM_XCC, &xcc);
          g_core.hsa_agent_get_info_fn(
              agent, (hsa_agent_info_t) HSA_AMD_AGENT_INFO_DRIVER_NODE_ID, &node);
          rec.node_id = node;
          rec.wave_front_size = wfs;
          rec.cu_count = cus;
          rec.num_xcc = xcc;
          g_core.hsa_agent_get_info_fn(agent, HSA_AGENT_INFO_NAME, rec.name);
          rec.name[sizeof(rec.name) - 1] = 0;
          write_raw(&rec, sizeof(rec));
        }
        agents_map()[agent.handle] = info;
        return HSA_STATUS_SUCCESS;
      },
      &ctx);
  g_agents_done.store(true);
}

int32_t agent_device(hsa_agent_t a) {
  auto& m = agents_map();
  auto it = m.find(a.handle);
  return it == m.end() ? -1 : it->second.device;
}

// ------------------------------------------------------- per-queue dispatch

// One pending in-flight dispatch awaiting its GPU-only timing signal.
struct PendSlot {
  hsa_signal_t sig{};       // ours (GPU-only, initial value 1)
  hsa_signal_t orig_sig{};  // app's signal if we replaced it (else 0)
  uint64_t kernel_object = 0;
  uint32_t tid = 0;
  uint32_t private_segment_size = 0;
  uint32_t group_segment_size = 0;
  uint16_t wg[3] = {0, 0, 0};
  uint32_t grid[3] = {0, 0, 0};
};

struct QueueCtx {
  hsa_queue_t* queue = nullptr;
  hsa_agent_t agent{};
  int32_t device = 0;
  uint64_t queue_id = 0;
  std::atomic<bool> alive{true};

  // slot pool + in-flight list (reaper swaps the in-flight vector out)
  pthread_spinlock_t lock;
  std::vector<PendSlot> slots;
  std::vector<uint32_t> free_idx;
  std::vector<uint32_t> inflight;

  QueueCtx() { pthread_spin_init(&lock, PTHREAD_PROCESS_PRIVATE); }
};

std::mutex g_queues_mutex;
std::vector<QueueCtx*>& queues_vec() {  // leaked (see agents_map comment)
  static auto* v = new std::vector<QueueCtx*>();
  return *v;
}

constexpr uint32_t kSlotsPerQueue = 8192;

// ApiRec op id for host-side AQL submit spans (preprocess resolves the name
// through an OpNameRec written at init)
constexpr uint32_t kAqlSubmitOp = 60000;
bool g_submit_spans = true;  // SOFA_LITE_SUBMIT_SPANS=0 to disable
// SOFA_LITE_SAMPLE=N: attach a timing signal to every Nth dispatch only
// (1 = every dispatch, the default).  Overhead scales ~1/N while the
// submit spans still record every launch — the always-on production dial.
uint32_t g_sample_every = 1;
std::atomic<uint64_t> g_sample_counter{0};

std::atomic<uint64_t> g_pool_exhausted{0};

// diagnostics (SOFA_LITE_DEBUG=1 prints the full set at exit)
struct Stats {
  std::atomic<uint64_t> queue_create_gpu{0};
  std::atomic<uint64_t> submit_batches{0};
  std::atomic<uint64_t> kernel_pkts{0};
  std::atomic<uint64_t> attached{0};
  std::atomic<uint64_t> skipped_has_signal{0};
  std::atomic<uint64_t> reaped{0};
  std::atomic<uint64_t> copy_calls{0};
  std::atomic<uint64_t> copy_engine_calls{0};
  std::atomic<uint64_t> copy_rect_calls{0};
  std::atomic<uint64_t> copies_recorded{0};
} g_stats;

pthread_t g_reaper;
std::atomic<bool> g_reaper_started{false};

// forward decl
void* reaper_main(void*);

void ensure_reaper() {
  bool expected = false;
  if (g_reaper_started.compare_exchange_strong(expected, true)) {
    pthread_create(&g_reaper, nullptr, reaper_main, nullptr);
  }
}

// AQL header helpers
inline uint8_t packet_type(uint16_t header) { return header & 0xFF; }

// per-thread submit-span buffer: the submitting thread appends under a
// per-buffer spinlock (uncontended in steady state); the reaper swaps the
// vector out on its own cadence
struct SubmitBuf {
  pthread_spinlock_t lock;
  std::vector<sgt::ApiRec> recs;
  SubmitBuf() {
    pthread_spin_init(&lock, PTHREAD_PROCESS_PRIVATE);
    recs.reserve(1024);
  }
};
std::mutex g_subufs_mutex;
std::vector<SubmitBuf*>& subufs_vec() {  // leaked (see agents_map comment)
  static auto* v = new std::vector<SubmitBuf*>();
  return *v;
}

void submit_buf_push(const sgt::ApiRec& rec) {
  static thread_local SubmitBuf* buf = [] {
    auto* b = new SubmitBuf();
    std::lock_guard<std::mutex> lk(g_subufs_mutex);
    subufs_vec().push_back(b);
    return b;
  }();
  pthread_spin_lock(&buf->lock);
  buf->recs.push_back(rec);
  pthread_spin_unlock(&buf->lock);
}

void drain_submit_bufs(std::vector<char>& chunk) {
  std::vector<SubmitBuf*> bufs;
  {
    std::lock_guard<std::mutex> lk(g_subufs_mutex);
    bufs = subufs_vec();
  }
  for (auto* b : bufs) {
    pthread_spin_lock(&b->lock);
    std::vector<sgt::ApiRec> got;
    got.swap(b->recs);
    pthread_spin_unlock(&b->lock);
    if (!got.empty())
      chunk.insert(chunk.end(), (const char*) got.data(),
                   (const char*) (got.data() + got.size()));
  }
}

// The submit-side handler: the ONLY code on the application's critical path.
void on_submit(const void* pkts, uint64_t pkt_count, uint64_t /*user_pkt_index*/,
               void* data, hsa_amd_queue_intercept_packet_writer writer) {
  auto* q = static_cast<QueueCtx*>(data);
  if (g_mode != MODE_FULL || !g_armed.load(std::memory_order_relaxed) ||
      g_shutdown.load(std::memory_order_relaxed)) {
    writer(pkts, pkt_count);
    return;
  }
  g_stats.submit_batches.fetch_add(1, std::memory_order_relaxed);
  uint64_t submit_t0 = g_submit_spans ? sys_now_ns() : 0;
  // find kernel dispatch packets we can instrument
  const auto* in = static_cast<const hsa_kernel_dispatch_packet_t*>(pkts);
  static thread_local std::vector<hsa_kernel_dispatch_packet_t> scratch;
  bool patched = false;
  for (uint64_t i = 0; i < pkt_count; ++i) {
    uint8_t type = packet_type(in[i].header);
    if (type != HSA_PACKET_TYPE_KERNEL_DISPATCH) continue;
    g_stats.kernel_pkts.fetch_add(1, std::memory_order_relaxed);
    bool has_sig = in[i].completion_signal.handle != 0;
    if (has_sig && !g_replace_signals) {
      g_stats.skipped_has_signal.fetch_add(1, std::memory_order_relaxed);
      continue;
    }
    if (g_sample_every > 1 &&
        (g_sample_counter.fetch_add(1, std::memory_order_relaxed) %
         g_sample_every) != 0)
      continue;  // sampled-timing mode: this dispatch goes untimed

    // acquire a slot + publish to the in-flight list in ONE critical
    // section (an unpublished-packet slot is harmless to the reaper: its
    // signal still reads 1, so it just stays pending) — halves the lock
    // traffic on the hot submit path
    pthread_spin_lock(&q->lock);
    if (q->free_idx.empty()) {
      pthread_spin_unlock(&q->lock);
      g_pool_exhausted.fetch_add(1, std::memory_order_relaxed);
      continue;  // forward untimed rather than stall the app
    }
    uint32_t idx = q->free_idx.back();
    q->free_idx.pop_back();
    q->inflight.push_back(idx);
    pthread_spin_unlock(&q->lock);

    PendSlot& s = q->slots[idx];
    s.orig_sig = has_sig ? in[i].completion_signal : hsa_signal_t{0};
    s.kernel_object = in[i].kernel_object;
    s.tid = my_tid();
    s.private_segment_size = in[i].private_segment_size;
    s.group_segment_size = in[i].group_segment_size;
    s.wg[0] = in[i].workgroup_size_x;
    s.wg[1] = in[i].workgroup_size_y;
    s.wg[2] = in[i].workgroup_size_z;
    s.grid[0] = in[i].grid_size_x;
    s.grid[1] = in[i].grid_size_y;
    s.grid[2] = in[i].grid_size_z;

    if (!patched) {
      scratch.assign(in, in + pkt_count);
      patched = true;
    }
    scratch[i].completion_signal = s.sig;
    g_stats.attached.fetch_add(1, std::memory_order_relaxed);
  }
  writer(patched ? (const void*) scratch.data() : pkts, pkt_count);
  if (g_submit_spans) {
    // thread-local buffering: no lock, no fwrite on the submit path (the
    // global write mutex contends with the reaper's flushes — measured
    // ~0.5% of step time when written inline)
    sgt::ApiRec rec{};
    rec.h = {sgt::REC_HIPAPI, sizeof(sgt::ApiRec), 0};
    rec.start_ns = submit_t0;
    rec.end_ns = sys_now_ns();
    rec.corr_id = pkt_count;  // packets in this batch
    rec.tid = my_tid();
    rec.op = kAqlSubmitOp;
    submit_buf_push(rec);
    g_n_records.fetch_add(1, std::memory_order_relaxed);
  }
}

// Reaper: poll in-flight slots; on completion read HW timestamps, emit
// KernelRec, recycle the signal.  Runs entirely off the app's critical path.
void* reaper_main(void*) {
  pthread_setname_np(pthread_self(), "sofa-lite-reap");
  std::vector<char> chunk;
  std::vector<uint32_t> still;
  while (!g_shutdown.load(std::memory_order_acquire)) {
    bool any_inflight = false;
    size_t n_reaped = 0;
    size_t nq = 0;
    {
      std::lock_guard<std::mutex> lk(g_queues_mutex);
      nq = queues_vec().size();
    }
    for (size_t qi = 0; qi < nq; ++qi) {
      QueueCtx* q;
      {
        std::lock_guard<std::mutex> lk(g_queues_mutex);
        q = queues_vec()[qi];
      }
      // swap out the inflight list
      pthread_spin_lock(&q->lock);
      std::vector<uint32_t> work;
      work.swap(q->inflight);
      pthread_spin_unlock(&q->lock);
      if (work.empty()) continue;
      still.clear();
      for (uint32_t idx : work) {
        PendSlot& s = q->slots[idx];
        hsa_signal_value_t v = g_core.hsa_signal_load_scacquire_fn(s.sig);
        if (v > 0) {
          still.push_back(idx);
          continue;
        }
        // completed: read CP timestamps
        hsa_amd_profiling_dispatch_time_t t{};
        hsa_status_t st =
            g_amd.hsa_amd_profiling_get_dispatch_time_fn(q->agent, s.sig, &t);
        if (s.orig_sig.handle != 0) {
          g_core.hsa_signal_subtract_screlease_fn(s.orig_sig, 1);
        }
        if (st == HSA_STATUS_SUCCESS && t.end >= t.start) {
          sgt::KernelRec rec{};
          rec.h = {sgt::REC_KERNEL, sizeof(sgt::KernelRec), 0};
          rec.start_ns = (uint64_t) (t.start * g_tick_to_ns);
          rec.end_ns = (uint64_t) (t.end * g_tick_to_ns);
          rec.corr_id = 0;
          rec.tid = s.tid;
          rec.device = (uint32_t) (q->device < 0 ? 0 : q->device);
          rec.queue_id = q->queue_id;
          rec.kernel_id = s.kernel_object;
          rec.private_segment_size = s.private_segment_size;
          rec.group_segment_size = s.group_segment_size;
          rec.grid_x = s.grid[0];
          rec.grid_y = s.grid[1];
          rec.grid_z = s.grid[2];
          rec.wg_x = s.wg[0];
          rec.wg_y = s.wg[1];
          rec.wg_z = s.wg[2];
          chunk.insert(chunk.end(), (const char*) &rec,
                       (const char*) &rec + sizeof(rec));
          ++n_reaped;
          g_stats.reaped.fetch_add(1, std::memory_order_relaxed);
        }
        // recycle
        g_core.hsa_signal_store_screlease_fn(s.sig, 1);
        s.orig_sig.handle = 0;
        pthread_spin_lock(&q->lock);
        q->free_idx.push_back(idx);
        pthread_spin_unlock(&q->lock);
      }
      if (!still.empty()) {
        any_inflight = true;
        pthread_spin_lock(&q->lock);
        // new submissions may have appended meanwhile; merge
        for (uint32_t idx : still) q->inflight.push_back(idx);
        pthread_spin_unlock(&q->lock);
      }
    }
    drain_submit_bufs(chunk);
    if (!chunk.empty()) {
      write_raw(chunk.data(), chunk.size());
      g_n_records.fetch_add(n_reaped, std::memory_order_relaxed);
      chunk.clear();
    }
    // adaptive cadence: busy-ish while work is pending, sleepy when idle
    struct timespec ts {0, any_inflight ? 20'000 : 200'000};
    nanosleep(&ts, nullptr);
  }
  drain_submit_bufs(chunk);
  if (!chunk.empty()) write_raw(chunk.data(), chunk.size());
  return nullptr;
}

// ------------------------------------------------------------ API wrappers

hsa_status_t queue_create_wrap(hsa_agent_t agent, uint32_t size,
                               hsa_queue_type32_t type,
                               void (*callback)(hsa_status_t, hsa_queue_t*, void*),
                               void* data, uint32_t private_segment_size,
                               uint32_t group_segment_size, hsa_queue_t** queue) {
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  enumerate_agents();
  hsa_device_type_t dev_type = HSA_DEVICE_TYPE_CPU;
  g_core.hsa_agent_get_info_fn(agent, HSA_AGENT_INFO_DEVICE, &dev_type);
  if (dev_type == HSA_DEVICE_TYPE_GPU)
    g_stats.queue_create_gpu.fetch_add(1, std::memory_order_relaxed);
  if (dev_type != HSA_DEVICE_TYPE_GPU || g_mode == MODE_OFF || !g_out) {
    // CPU soft queues / disabled: plain queue (MODE_OFF still proxies below
    // to measure the proxy floor — but with no handler registered)
    if (dev_type != HSA_DEVICE_TYPE_GPU || !g_out)
      return g_core.hsa_queue_create_fn(agent, size, type, callback, data,
                                        private_segment_size,
                                        group_segment_size, queue);
  }
  hsa_status_t st = g_amd.hsa_amd_queue_intercept_create_fn(
      agent, size, type, callback, data, private_segment_size,
      group_segment_size, queue);
  if (st != HSA_STATUS_SUCCESS) return st;

  if (g_mode >= MODE_PROF) {
    g_amd.hsa_amd_profiling_set_profiler_enabled_fn(*queue, 1);
  }
  if (g_mode == MODE_FULL) {
    auto* q = new QueueCtx();
    q->queue = *queue;
    q->agent = agent;
    q->device = agent_device(agent);
    q->queue_id = (*queue)->id;
    q->slots.resize(kSlotsPerQueue);
    q->free_idx.reserve(kSlotsPerQueue);
    bool ok = true;
    for (uint32_t i = 0; i < kSlotsPerQueue; ++i) {
      // GPU-only: CP writes timestamps + decrements with NO host interrupt
      if (g_amd.hsa_amd_signal_create_fn(1, 0, nullptr,
                                         HSA_AMD_SIGNAL_AMD_GPU_ONLY,
                                         &q->slots[i].sig) !=
          HSA_STATUS_SUCCESS) {
        ok = false;
        break;
      }
      q->free_idx.push_back(kSlotsPerQueue - 1 - i);
    }
    if (ok) {
      g_amd.hsa_amd_queue_intercept_register_fn(*queue, on_submit, q);
      {
        std::lock_guard<std::mutex> lk(g_queues_mutex);
        queues_vec().push_back(q);
      }
      ensure_reaper();
    } else {
      delete q;
    }
  }
  return st;
}

void finalize_for_shutdown();

hsa_status_t hsa_shut_down_wrap() {
  // stop the reaper and close the file while signals are still alive; ROCr
  // refcounts init/shutdown but one shutdown from the app is the exit path
  if (env_flag("SOFA_LITE_DEBUG", false))
    fprintf(stderr, "[sofahsalite] hsa_shut_down intercepted\n");
  finalize_for_shutdown();
  g_runtime_down.store(true, std::memory_order_release);
  return g_core.hsa_shut_down_fn();
}

hsa_status_t queue_destroy_wrap(hsa_queue_t* queue) {
  if (env_flag("SOFA_LITE_DEBUG", false))
    fprintf(stderr, "[sofahsalite] queue_destroy %p (down=%d)\n",
            (void*) queue, (int) g_runtime_down.load());
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  // drain: reaper keeps polling; just mark dead and let slots finish.
  // ROCr destroys the proxy after in-flight packets retire, so pending
  // signals have fired by then; a short grace wait covers the reap gap.
  {
    std::lock_guard<std::mutex> lk(g_queues_mutex);
    for (auto* q : queues_vec())
      if (q->queue == queue) q->alive.store(false);
  }
  for (int spin = 0; spin < 50; ++spin) {
    bool pending = false;
    {
      std::lock_guard<std::mutex> lk(g_queues_mutex);
      for (auto* q : queues_vec())
        if (q->queue == queue) {
          pthread_spin_lock(&q->lock);
          pending = !q->inflight.empty();
          pthread_spin_unlock(&q->lock);
        }
    }
    if (!pending) break;
    usleep(1000);
  }
  return g_core.hsa_queue_destroy_fn(queue);
}

// ---- kernel symbol names: executable freeze -> NameRec(kernel_object)

hsa_status_t exe_freeze_wrap(hsa_executable_t executable, const char* options) {
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  hsa_status_t st = g_core.hsa_executable_freeze_fn(executable, options);
  if (st != HSA_STATUS_SUCCESS || !g_out) return st;
  g_core.hsa_executable_iterate_symbols_fn(
      executable,
      [](hsa_executable_t, hsa_executable_symbol_t sym, void*) -> hsa_status_t {
        hsa_symbol_kind_t kind = HSA_SYMBOL_KIND_VARIABLE;
        g_core.hsa_executable_symbol_get_info_fn(
            sym, HSA_EXECUTABLE_SYMBOL_INFO_TYPE, &kind);
        if (kind != HSA_SYMBOL_KIND_KERNEL) return HSA_STATUS_SUCCESS;
        uint64_t kobj = 0;
        g_core.hsa_executable_symbol_get_info_fn(
            sym, HSA_EXECUTABLE_SYMBOL_INFO_KERNEL_OBJECT, &kobj);
        uint32_t len = 0;
        g_core.hsa_executable_symbol_get_info_fn(
            sym, HSA_EXECUTABLE_SYMBOL_INFO_NAME_LENGTH, &len);
        if (!kobj || !len || len > kMaxNameLen) return HSA_STATUS_SUCCESS;
        std::vector<char> name(len + 1, 0);
        g_core.hsa_executable_symbol_get_info_fn(
            sym, HSA_EXECUTABLE_SYMBOL_INFO_NAME, name.data());
        // HSA kernel symbols carry a ".kd" suffix the demangler chokes on
        if (len > 3 && memcmp(name.data() + len - 3, ".kd", 3) == 0) len -= 3;
        write_name_rec(sgt::REC_KERNEL_NAME, kobj, name.data(), len);
        return HSA_STATUS_SUCCESS;
      },
      nullptr);
  return st;
}

// ---- async SDMA copies

struct CopySlot {
  hsa_signal_t sig{};       // ours, interrupt-capable (async handler)
  hsa_signal_t orig_sig{};  // caller's completion signal
  uint64_t bytes = 0;
  int32_t src_device = -1;
  int32_t dst_device = -1;
  uint32_t op = 0;  // rocprofiler_memory_copy_operation_t codes 1..4
  uint32_t tid = 0;
};

std::mutex g_copy_mutex;
std::vector<CopySlot*>& copy_free_vec() {  // leaked (see agents_map comment)
  static auto* v = new std::vector<CopySlot*>();
  return *v;
}
std::atomic<bool> g_copy_prof_enabled{false};

// hsa_amd_profiling_async_copy_enable cannot run from OnLoad (mid-hsa_init);
// enable lazily at the first wrapped copy call
void ensure_copy_profiling() {
  bool expected = false;
  if (g_copy_prof_enabled.compare_exchange_strong(expected, true)) {
    hsa_status_t st = g_amd.hsa_amd_profiling_async_copy_enable_fn(true);
    if (st != HSA_STATUS_SUCCESS && env_flag("SOFA_LITE_DEBUG", false))
      fprintf(stderr, "[sofahsalite] async_copy_enable failed: %d\n", st);
  }
}

CopySlot* copy_slot_get() {
  std::lock_guard<std::mutex> lk(g_copy_mutex);
  auto& v = copy_free_vec();
  if (!v.empty()) {
    CopySlot* s = v.back();
    v.pop_back();
    return s;
  }
  auto* s = new CopySlot();
  if (g_core.hsa_signal_create_fn(1, 0, nullptr, &s->sig) != HSA_STATUS_SUCCESS) {
    delete s;
    return nullptr;
  }
  return s;
}

void copy_slot_put(CopySlot* s) {
  g_core.hsa_signal_store_screlease_fn(s->sig, 1);
  std::lock_guard<std::mutex> lk(g_copy_mutex);
  copy_free_vec().push_back(s);
}

std::atomic<uint64_t> g_copy_handler_fired{0};
std::atomic<uint64_t> g_copy_time_fail{0};

bool copy_done_handler(hsa_signal_value_t, void* arg) {
  g_copy_handler_fired.fetch_add(1, std::memory_order_relaxed);
  auto* s = static_cast<CopySlot*>(arg);
  hsa_amd_profiling_async_copy_time_t t{};
  hsa_status_t st = g_amd.hsa_amd_profiling_get_async_copy_time_fn(s->sig, &t);
  if (st != HSA_STATUS_SUCCESS)
    g_copy_time_fail.fetch_add(1, std::memory_order_relaxed);
  // forward the caller's completion FIRST (its waiters matter more than our
  // record)
  if (s->orig_sig.handle != 0)
    g_core.hsa_signal_subtract_screlease_fn(s->orig_sig, 1);
  if (st == HSA_STATUS_SUCCESS && t.end >= t.start) {
    sgt::CopyRec rec{};
    rec.h = {sgt::REC_COPY, sizeof(sgt::CopyRec), 0};
    rec.start_ns = (uint64_t) (t.start * g_tick_to_ns);
    rec.end_ns = (uint64_t) (t.end * g_tick_to_ns);
    rec.corr_id = 0;
    rec.tid = s->tid;
    rec.op = s->op;
    rec.src_device = s->src_device;
    rec.dst_device = s->dst_device;
    rec.bytes = s->bytes;
    write_raw(&rec, sizeof(rec));
    g_n_records.fetch_add(1, std::memory_order_relaxed);
    g_stats.copies_recorded.fetch_add(1, std::memory_order_relaxed);
  }
  copy_slot_put(s);
  return false;  // one-shot
}

hsa_status_t async_copy_wrap(void* dst, hsa_agent_t dst_agent, const void* src,
                             hsa_agent_t src_agent, size_t size,
                             uint32_t num_dep_signals,
                             const hsa_signal_t* dep_signals,
                             hsa_signal_t completion_signal) {
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  g_stats.copy_calls.fetch_add(1, std::memory_order_relaxed);
  // ROCclr passes completion_signal.handle == 0 on its SDMA path (measured
  // on MI355X); a zero signal means nobody waits on it, so attaching ours is
  // free of semantic risk — same rule as kernel packets.
  if (g_mode != MODE_FULL || !g_armed.load(std::memory_order_relaxed)) {
    return g_amd.hsa_amd_memory_async_copy_fn(dst, dst_agent, src, src_agent,
                                              size, num_dep_signals,
                                              dep_signals, completion_signal);
  }
  enumerate_agents();
  ensure_copy_profiling();
  CopySlot* s = copy_slot_get();
  if (!s)
    return g_amd.hsa_amd_memory_async_copy_fn(dst, dst_agent, src, src_agent,
                                              size, num_dep_signals,
                                              dep_signals, completion_signal);
  s->orig_sig = completion_signal;
  s->bytes = size;
  s->src_device = agent_device(src_agent);
  s->dst_device = agent_device(dst_agent);
  bool src_gpu = s->src_device >= 0, dst_gpu = s->dst_device >= 0;
  s->op = (!src_gpu && !dst_gpu) ? 1 : (!src_gpu && dst_gpu) ? 2
          : (src_gpu && !dst_gpu) ? 3 : 4;
  s->tid = my_tid();
  hsa_status_t st = g_amd.hsa_amd_memory_async_copy_fn(
      dst, dst_agent, src, src_agent, size, num_dep_signals, dep_signals,
      s->sig);
  if (st != HSA_STATUS_SUCCESS) {
    copy_slot_put(s);
    return st;
  }
  // register AFTER the successful submit: an already-satisfied condition
  // fires the handler immediately, so no completion can be missed
  g_amd.hsa_amd_signal_async_handler_fn(s->sig, HSA_SIGNAL_CONDITION_LT, 1,
                                        copy_done_handler, s);
  return st;
}

hsa_status_t async_copy_engine_wrap(void* dst, hsa_agent_t dst_agent,
                                    const void* src, hsa_agent_t src_agent,
                                    size_t size, uint32_t num_dep_signals,
                                    const hsa_signal_t* dep_signals,
                                    hsa_signal_t completion_signal,
                                    hsa_amd_sdma_engine_id_t engine_id,
                                    bool force_copy_on_sdma) {
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  g_stats.copy_engine_calls.fetch_add(1, std::memory_order_relaxed);
  if (g_mode != MODE_FULL || !g_armed.load(std::memory_order_relaxed)) {
    return g_amd.hsa_amd_memory_async_copy_on_engine_fn(
        dst, dst_agent, src, src_agent, size, num_dep_signals, dep_signals,
        completion_signal, engine_id, force_copy_on_sdma);
  }
  enumerate_agents();
  ensure_copy_profiling();
  CopySlot* s = copy_slot_get();
  if (!s)
    return g_amd.hsa_amd_memory_async_copy_on_engine_fn(
        dst, dst_agent, src, src_agent, size, num_dep_signals, dep_signals,
        completion_signal, engine_id, force_copy_on_sdma);
  s->orig_sig = completion_signal;
  s->bytes = size;
  s->src_device = agent_device(src_agent);
  s->dst_device = agent_device(dst_agent);
  bool src_gpu = s->src_device >= 0, dst_gpu = s->dst_device >= 0;
  s->op = (!src_gpu && !dst_gpu) ? 1 : (!src_gpu && dst_gpu) ? 2
          : (src_gpu && !dst_gpu) ? 3 : 4;
  s->tid = my_tid();
  hsa_status_t st = g_amd.hsa_amd_memory_async_copy_on_engine_fn(
      dst, dst_agent, src, src_agent, size, num_dep_signals, dep_signals,
      s->sig, engine_id, force_copy_on_sdma);
  if (st != HSA_STATUS_SUCCESS) {
    copy_slot_put(s);
    return st;
  }
  g_amd.hsa_amd_signal_async_handler_fn(s->sig, HSA_SIGNAL_CONDITION_LT, 1,
                                        copy_done_handler, s);
  return st;
}

hsa_status_t async_copy_rect_wrap(const hsa_pitched_ptr_t* dst,
                                  const hsa_dim3_t* dst_offset,
                                  const hsa_pitched_ptr_t* src,
                                  const hsa_dim3_t* src_offset,
                                  const hsa_dim3_t* range,
                                  hsa_agent_t copy_agent,
                                  hsa_amd_copy_direction_t dir,
                                  uint32_t num_dep_signals,
                                  const hsa_signal_t* dep_signals,
                                  hsa_signal_t completion_signal) {
  if (g_runtime_down.load(std::memory_order_acquire))
    return HSA_STATUS_ERROR_NOT_INITIALIZED;
  g_stats.copy_rect_calls.fetch_add(1, std::memory_order_relaxed);
  return g_amd.hsa_amd_memory_async_copy_rect_fn(
      dst, dst_offset, src, src_offset, range, copy_agent, dir,
      num_dep_signals, dep_signals, completion_signal);
}

// ------------------------------------------------------------------ control

void open_output() {
  const char* logdir = getenv("SOFA_LOGDIR");
  if (!logdir || !*logdir) logdir = ".";
  char path[4096];
  snprintf(path, sizeof(path), "%s/gputrace_%d_lite.sgt", logdir, getpid());
  g_out = fopen(path, "wb");
  if (!g_out) {
    fprintf(stderr, "[sofahsalite] cannot open %s\n", path);
    return;
  }
  uint64_t freq = 0;
  g_core.hsa_system_get_info_fn(HSA_SYSTEM_INFO_TIMESTAMP_FREQUENCY, &freq);
  if (freq > 0) g_tick_to_ns = 1e9 / (double) freq;

  sgt::FileHeader hdr{};
  hdr.magic = sgt::kMagic;
  hdr.version = sgt::kVersion;
  hdr.pid = (uint32_t) getpid();
  hdr.realtime_ns = host_ns(CLOCK_REALTIME);
  hdr.monotonic_raw_ns = host_ns(CLOCK_MONOTONIC_RAW);
  hdr.rocp_ns = sys_now_ns();
  fwrite(&hdr, sizeof(hdr), 1, g_out);
  write_opname_rec(3 /* HIP_RUNTIME_API kind */, 60000, "aqlSubmitBatch");
  write_clock_rec();
}

void finalize();
void finalize_for_shutdown() { finalize(); }

void finalize() {
  static std::atomic<bool> done{false};
  bool expected = false;
  if (!done.compare_exchange_strong(expected, true)) return;
  g_shutdown.store(true, std::memory_order_release);
  if (g_reaper_started.load()) pthread_join(g_reaper, nullptr);
  // destroy every pooled signal BEFORE the runtime shuts down: leaked
  // GPU-only signals crash ROCr teardown (measured: lite-full exits 139,
  // lite-prof/off exit 0 — the only delta is the signal pools)
  {
    std::lock_guard<std::mutex> lk(g_queues_mutex);
    for (auto* q : queues_vec()) {
      for (auto& s : q->slots) {
        if (s.sig.handle) {
          g_core.hsa_signal_destroy_fn(s.sig);
          s.sig.handle = 0;
        }
      }
    }
  }
  {
    std::lock_guard<std::mutex> lk(g_copy_mutex);
    for (auto* s : copy_free_vec()) {
      if (s->sig.handle) {
        g_core.hsa_signal_destroy_fn(s->sig);
        s->sig.handle = 0;
      }
    }
  }
  if (env_flag("SOFA_LITE_DEBUG", false)) {
    fprintf(stderr,
            "[sofahsalite] queues=%lu batches=%lu kernel_pkts=%lu attached=%lu "
            "skipped_has_signal=%lu reaped=%lu copy=%lu copy_engine=%lu "
            "copy_rect=%lu copies_recorded=%lu pool_exhausted=%lu\n",
            g_stats.queue_create_gpu.load(), g_stats.submit_batches.load(),
            g_stats.kernel_pkts.load(), g_stats.attached.load(),
            g_stats.skipped_has_signal.load(), g_stats.reaped.load(),
            g_stats.copy_calls.load(), g_stats.copy_engine_calls.load(),
            g_stats.copy_rect_calls.load(), g_stats.copies_recorded.load(),
            g_pool_exhausted.load());
    fprintf(stderr, "[sofahsalite] copy_handler_fired=%lu copy_time_fail=%lu\n",
            g_copy_handler_fired.load(), g_copy_time_fail.load());
  }
  if (g_out) {
    write_clock_rec();
    if (g_pool_exhausted.load() > 0) {
      sgt::DropRec d{};
      d.h = {sgt::REC_DROP, sizeof(sgt::DropRec), 0};
      d.dropped = g_pool_exhausted.load();
      write_raw(&d, sizeof(d));
    }
    std::lock_guard<std::mutex> lk(g_out_mutex);
    fclose(g_out);
    g_out = nullptr;
  }
}

}  // namespace

extern "C" {

// bench.py A/B control (mirrors sofa_tracer_start/stop)
int sofa_lite_start() {
  g_armed.store(true, std::memory_order_relaxed);
  return 0;
}

int sofa_lite_stop() {
  g_armed.store(false, std::memory_order_relaxed);
  // give the reaper one cycle to drain records from the stopped phase
  usleep(2000);
  return 0;
}

unsigned long long sofa_lite_event_count() {
  return g_n_records.load(std::memory_order_relaxed);
}

int sofa_lite_flush() {
  std::lock_guard<std::mutex> lk(g_out_mutex);
  if (g_out) fflush(g_out);
  return 0;
}

int sofa_lite_active() { return g_out != nullptr; }

unsigned long long sofa_lite_dropped() { return g_pool_exhausted.load(); }

// SOFA_LITE_DEBUG=1 diagnostics: constructor fires iff ROCr dlopens us,
// OnLoad print fires iff the tools ABI call happens — separates "never
// loaded" from "loaded but not initialized" when other tools are present.
__attribute__((constructor)) static void sofa_lite_ctor() {
  if (env_flag("SOFA_LITE_DEBUG", false))
    fprintf(stderr, "[sofahsalite] dlopened (pid %d)\n", getpid());
}

// ROCr tools-library entry points (HSA_TOOLS_LIB)
bool OnLoad(void* table_v, uint64_t runtime_version, uint64_t failed_tool_count,
            const char* const* failed_tool_names) {
  if (env_flag("SOFA_LITE_DEBUG", false))
    fprintf(stderr, "[sofahsalite] OnLoad (runtime_version %lu)\n",
            (unsigned long) runtime_version);
  (void) runtime_version;
  (void) failed_tool_count;
  (void) failed_tool_names;
  auto* table = static_cast<HsaApiTable*>(table_v);
  if (!table || !table->core_ || !table->amd_ext_) return true;
  g_core = *table->core_;
  g_amd = *table->amd_ext_;
  g_saved = true;

  const char* mode = getenv("SOFA_LITE_MODE");
  g_mode = MODE_FULL;
  if (mode) {
    if (!strcmp(mode, "off")) g_mode = MODE_OFF;
    else if (!strcmp(mode, "prof")) g_mode = MODE_PROF;
  }
  g_replace_signals = env_flag("SOFA_LITE_REPLACE_SIGNALS", false);
  g_submit_spans = env_flag("SOFA_LITE_SUBMIT_SPANS", true);
  if (const char* v = getenv("SOFA_LITE_SAMPLE"); v && *v) {
    unsigned long n = strtoul(v, nullptr, 10);
    if (n >= 1) g_sample_every = (uint32_t) n;
  }
  g_armed.store(!env_flag("SOFA_DEFER_START", false));

  open_output();
  if (!g_out) return true;  // stay passive

  table->core_->hsa_queue_create_fn = queue_create_wrap;
  table->core_->hsa_queue_destroy_fn = queue_destroy_wrap;
  table->core_->hsa_executable_freeze_fn = exe_freeze_wrap;
  table->core_->hsa_shut_down_fn = hsa_shut_down_wrap;
  if (g_mode == MODE_FULL) {
    table->amd_ext_->hsa_amd_memory_async_copy_fn = async_copy_wrap;
    table->amd_ext_->hsa_amd_memory_async_copy_on_engine_fn =
        async_copy_engine_wrap;
    table->amd_ext_->hsa_amd_memory_async_copy_rect_fn = async_copy_rect_wrap;
  }
  atexit(finalize);
  return true;
}

void OnUnload() { finalize(); }

}  // extern "C"
