// libsofatracer — rocprofiler-sdk tool library: the MI355X-native replacement
// for the reference's system-wide nvprof/CUPTI capture
// (cyliustack/sofa bin/sofa_record.py:217-242 + nvvp/sqlite parsing in
// bin/sofa_preprocess.py:1340-1543).
//
// Loaded into every GPU process of the profiled command via
// ROCP_TOOL_LIBRARIES (rocprofiler-register picks it up when the HIP/HSA
// runtime initializes).  Captures, with buffered (lossless) tracing:
//   * kernel dispatches (agent, queue, kernel_id, grid/workgroup, LDS/scratch)
//   * async memory copies (H2D/D2H/D2D/P2P with src/dst device)
//   * HIP runtime API spans (optional, SOFA_TRACE_HIP_API=1)
//   * memory allocations (optional)
// and with callback tracing:
//   * RCCL API calls with full args (count, dtype, comm, stream, peer/root) —
//     the basis for per-xGMI-link collective attribution in sofa_analyze
//   * code-object kernel-symbol registration (kernel_id -> mangled name)
//   * roctx markers (MARKER_CORE API) for user annotations.
//
// Output: one binary SGT file per process in $SOFA_LOGDIR (sgt_format.h);
// parsed vectorized (numpy structured dtypes) by sofa_amd.preprocess.gpu.
// Timestamps are rocprofiler's ns clock; REC_CLOCK records at init+fini give
// the (REALTIME, MONOTONIC_RAW, rocp) correlation used by preprocess to place
// GPU events on the unified timeline (SURVEY.md §7 "three-clock sync").
//
// Env knobs:
//   SOFA_LOGDIR        output directory (default: cwd)
//   SOFA_TRACE_HIP_API 1/0 (default 1)
//   SOFA_TRACE_RCCL    1/0 (default 1)
//   SOFA_TRACE_ALLOC   1/0 (default 0)
//   SOFA_GPU_BUFFER_MB per-process SDK buffer MiB (default 64)

#include <rocprofiler-sdk/pc_sampling.h>
#include <rocprofiler-sdk/registration.h>
#include <rocprofiler-sdk/rocprofiler.h>
#include <rocprofiler-sdk/rccl.h>
#include <rocprofiler-sdk/marker/api_id.h>

#include <dlfcn.h>
#include <unistd.h>

#include <atomic>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <mutex>
#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

#include "sgt_format.h"

namespace {

rocprofiler_client_id_t* g_client_id = nullptr;
rocprofiler_context_id_t g_ctx = {0};
rocprofiler_buffer_id_t g_buffer = {};
rocprofiler_buffer_id_t g_pc_buffer = {};
std::vector<rocprofiler_agent_id_t> g_gpu_agents;  // fill at write_agents
bool g_pc_active = false;

FILE* g_out = nullptr;
std::mutex g_mutex;
std::atomic<uint64_t> g_n_records{0};

// agent handle -> logical GPU index (or -1 for CPU agents).  Leaked on
// purpose: tool-library statics are destructed before the runtimes stop
// calling back (same hazard fixed in hsalite.cc).
std::unordered_map<uint64_t, int32_t>& agent_device_map() {
  static auto* m = new std::unordered_map<uint64_t, int32_t>();
  return *m;
}

bool env_flag(const char* name, bool dflt) {
  const char* v = getenv(name);
  if (!v || !*v) return dflt;
  return !(v[0] == '0' || v[0] == 'n' || v[0] == 'N' || v[0] == 'f' || v[0] == 'F');
}

uint64_t host_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

bool g_null_sink = false;  // SOFA_NULL_SINK=1: intercept but discard (the
                           // SDK-interception floor for overhead attribution)

void write_raw(const void* p, size_t n) {
  if (g_null_sink) return;
  std::lock_guard<std::mutex> lk(g_mutex);
  if (g_out) fwrite(p, 1, n, g_out);
}

// RecHeader.size is u16: clamp names so `total` cannot wrap the length
// prefix and misalign every subsequent record (round-1 ADVICE).
constexpr size_t kMaxNameLen = 65000;

void write_name_rec(uint16_t type, uint64_t id, const char* name) {
  size_t len = name ? strlen(name) : 0;
  if (len > kMaxNameLen) len = kMaxNameLen;
  size_t total = (sizeof(sgt::NameRec) + len + 1 + 7) & ~size_t(7);
  static_assert(sizeof(sgt::NameRec) + kMaxNameLen + 8 <= UINT16_MAX, "");
  std::vector<char> buf(total, 0);
  auto* rec = reinterpret_cast<sgt::NameRec*>(buf.data());
  rec->h = {type, static_cast<uint16_t>(total), 0};
  rec->id = id;
  if (len) memcpy(buf.data() + sizeof(sgt::NameRec), name, len);
  write_raw(buf.data(), total);
}

void write_opname_rec(uint32_t kind, uint32_t op, const char* name) {
  size_t len = name ? strlen(name) : 0;
  if (len > kMaxNameLen) len = kMaxNameLen;
  size_t total = (sizeof(sgt::OpNameRec) + len + 1 + 7) & ~size_t(7);
  std::vector<char> buf(total, 0);
  auto* rec = reinterpret_cast<sgt::OpNameRec*>(buf.data());
  rec->h = {sgt::REC_OPNAME, static_cast<uint16_t>(total), 0};
  rec->kind = kind;
  rec->op = op;
  if (len) memcpy(buf.data() + sizeof(sgt::OpNameRec), name, len);
  write_raw(buf.data(), total);
}

void write_clock_rec() {
  sgt::ClockRec rec{};
  rec.h = {sgt::REC_CLOCK, sizeof(sgt::ClockRec), 0};
  rocprofiler_timestamp_t ts = 0;
  rocprofiler_get_timestamp(&ts);
  rec.realtime_ns = host_ns(CLOCK_REALTIME);
  rec.monotonic_raw_ns = host_ns(CLOCK_MONOTONIC_RAW);
  rec.rocp_ns = ts;
  write_raw(&rec, sizeof(rec));
}

int32_t agent_device(rocprofiler_agent_id_t id) {
  auto& m = agent_device_map();
  auto it = m.find(id.handle);
  return it == m.end() ? -1 : it->second;
}

// RCCL datatype element sizes (ncclDataType_t order: int8, uint8, int32,
// uint32, int64, uint64, half, float, double, bf16, fp8e4m3, fp8e5m2)
uint32_t nccl_elem_size(uint32_t dt) {
  static const uint32_t sz[] = {1, 1, 4, 4, 8, 8, 2, 4, 8, 2, 1, 1};
  return dt < sizeof(sz) / sizeof(sz[0]) ? sz[dt] : 0;
}

// ---------------------------------------------------------------- callbacks

void code_object_callback(rocprofiler_callback_tracing_record_t record,
                          rocprofiler_user_data_t*, void*) {
  if (record.kind == ROCPROFILER_CALLBACK_TRACING_CODE_OBJECT &&
      record.operation == ROCPROFILER_CODE_OBJECT_DEVICE_KERNEL_SYMBOL_REGISTER &&
      record.phase == ROCPROFILER_CALLBACK_PHASE_LOAD) {
    auto* data = static_cast<
        rocprofiler_callback_tracing_code_object_kernel_symbol_register_data_t*>(
        record.payload);
    write_name_rec(sgt::REC_KERNEL_NAME, data->kernel_id, data->kernel_name);
  } else if (record.kind == ROCPROFILER_CALLBACK_TRACING_CODE_OBJECT &&
             record.operation == ROCPROFILER_CODE_OBJECT_LOAD &&
             record.phase == ROCPROFILER_CALLBACK_PHASE_UNLOAD) {
    auto status = rocprofiler_flush_buffer(g_buffer);
    (void) status;
  }
}

// Current HIP device of the calling thread, resolved lazily via dlsym so the
// tool library carries no link-time HIP dependency (it loads during runtime
// registration, before HIP is ready; the RCCL callback fires much later).
// Without this, every rank's collectives would report device 0 and per-link
// attribution in 8-GPU DDP (ranks pick GPUs via hipSetDevice, not
// HIP_VISIBLE_DEVICES) would pile onto one ring edge.
int current_hip_device() {
  using fn_t = int (*)(int*);
  static fn_t fn = reinterpret_cast<fn_t>(dlsym(RTLD_DEFAULT, "hipGetDevice"));
  int dev = 0;
  if (fn && fn(&dev) == 0) return dev;
  return 0;
}

// RCCL callback tracing: start/stop phases; we record the span + args.
struct RcclPending {
  uint64_t start_ns;
};
thread_local std::unordered_map<uint64_t, RcclPending> t_rccl_pending;

void rccl_callback(rocprofiler_callback_tracing_record_t record,
                   rocprofiler_user_data_t* user_data, void*) {
  if (record.kind != ROCPROFILER_CALLBACK_TRACING_RCCL_API) return;
  rocprofiler_timestamp_t now = 0;
  rocprofiler_get_timestamp(&now);
  if (record.phase == ROCPROFILER_CALLBACK_PHASE_ENTER) {
    user_data->value = now;
    return;
  }
  if (record.phase != ROCPROFILER_CALLBACK_PHASE_EXIT) return;

  auto* data =
      static_cast<rocprofiler_callback_tracing_rccl_api_data_t*>(record.payload);
  sgt::RcclRec rec{};
  rec.h = {sgt::REC_RCCL, sizeof(sgt::RcclRec), 0};
  rec.start_ns = user_data->value;
  rec.end_ns = now;
  rec.corr_id = record.correlation_id.internal;
  rec.tid = static_cast<uint32_t>(record.thread_id);
  rec.op = record.operation;
  rec.peer_or_root = -1;
  rec.device = (uint32_t) current_hip_device();

  const auto& a = data->args;
  switch (record.operation) {
    case ROCPROFILER_RCCL_API_ID_ncclAllReduce:
      rec.count = a.ncclAllReduce.count;
      rec.datatype = a.ncclAllReduce.datatype;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclAllReduce.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclAllReduce.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclAllGather:
      rec.count = a.ncclAllGather.sendcount;
      rec.datatype = a.ncclAllGather.datatype;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclAllGather.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclAllGather.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclReduceScatter:
      rec.count = a.ncclReduceScatter.recvcount;
      rec.datatype = a.ncclReduceScatter.datatype;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclReduceScatter.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclReduceScatter.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclAllToAll:
      rec.count = a.ncclAllToAll.count;
      rec.datatype = a.ncclAllToAll.datatype;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclAllToAll.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclAllToAll.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclBroadcast:
      rec.count = a.ncclBroadcast.count;
      rec.datatype = a.ncclBroadcast.datatype;
      rec.peer_or_root = a.ncclBroadcast.root;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclBroadcast.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclBroadcast.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclReduce:
      rec.count = a.ncclReduce.count;
      rec.datatype = a.ncclReduce.datatype;
      rec.peer_or_root = a.ncclReduce.root;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclReduce.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclReduce.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclSend:
      rec.count = a.ncclSend.count;
      rec.datatype = a.ncclSend.datatype;
      rec.peer_or_root = a.ncclSend.peer;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclSend.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclSend.stream);
      break;
    case ROCPROFILER_RCCL_API_ID_ncclRecv:
      rec.count = a.ncclRecv.count;
      rec.datatype = a.ncclRecv.datatype;
      rec.peer_or_root = a.ncclRecv.peer;
      rec.comm = reinterpret_cast<uint64_t>(a.ncclRecv.comm);
      rec.stream = reinterpret_cast<uint64_t>(a.ncclRecv.stream);
      break;
    default:
      break;
  }
  rec.elem_size = nccl_elem_size(rec.datatype);
  write_raw(&rec, sizeof(rec));
  g_n_records.fetch_add(1, std::memory_order_relaxed);
}

// roctx markers
void marker_callback(rocprofiler_callback_tracing_record_t record,
                     rocprofiler_user_data_t* user_data, void*) {
  if (record.kind != ROCPROFILER_CALLBACK_TRACING_MARKER_CORE_API) return;
  auto* data = static_cast<rocprofiler_callback_tracing_marker_api_data_t*>(
      record.payload);
  rocprofiler_timestamp_t now = 0;
  rocprofiler_get_timestamp(&now);
  if (record.phase == ROCPROFILER_CALLBACK_PHASE_ENTER) {
    if (record.operation == ROCPROFILER_MARKER_CORE_API_ID_roctxMarkA &&
        data->args.roctxMarkA.message) {
      write_name_rec(sgt::REC_MARKER, now, data->args.roctxMarkA.message);
    } else if (record.operation ==
                   ROCPROFILER_MARKER_CORE_API_ID_roctxRangePushA &&
               data->args.roctxRangePushA.message) {
      write_name_rec(sgt::REC_MARKER, now, data->args.roctxRangePushA.message);
    }
  }
  (void) user_data;
}

void buffer_callback(rocprofiler_context_id_t, rocprofiler_buffer_id_t,
                     rocprofiler_record_header_t** headers, size_t num_headers,
                     void*, uint64_t drop_count) {
  if (drop_count > 0) {
    sgt::DropRec d{};
    d.h = {sgt::REC_DROP, sizeof(sgt::DropRec), 0};
    d.dropped = drop_count;
    write_raw(&d, sizeof(d));
  }
  // Serialize the batch into one contiguous chunk, then one locked write.
  std::vector<char> chunk;
  chunk.reserve(num_headers * 80);
  auto emit = [&chunk](const void* p, size_t n) {
    const char* c = static_cast<const char*>(p);
    chunk.insert(chunk.end(), c, c + n);
  };
  for (size_t i = 0; i < num_headers; ++i) {
    auto* header = headers[i];
    if (header->category != ROCPROFILER_BUFFER_CATEGORY_TRACING) continue;
    switch (header->kind) {
      case ROCPROFILER_BUFFER_TRACING_KERNEL_DISPATCH: {
        auto* r = static_cast<rocprofiler_buffer_tracing_kernel_dispatch_record_t*>(
            header->payload);
        sgt::KernelRec rec{};
        rec.h = {sgt::REC_KERNEL, sizeof(sgt::KernelRec), 0};
        rec.start_ns = r->start_timestamp;
        rec.end_ns = r->end_timestamp;
        rec.corr_id = r->correlation_id.internal;
        rec.tid = static_cast<uint32_t>(r->thread_id);
        rec.device = static_cast<uint32_t>(
            agent_device(r->dispatch_info.agent_id) < 0
                ? 0
                : agent_device(r->dispatch_info.agent_id));
        rec.queue_id = r->dispatch_info.queue_id.handle;
        rec.kernel_id = r->dispatch_info.kernel_id;
        rec.private_segment_size = r->dispatch_info.private_segment_size;
        rec.group_segment_size = r->dispatch_info.group_segment_size;
        rec.grid_x = r->dispatch_info.grid_size.x;
        rec.grid_y = r->dispatch_info.grid_size.y;
        rec.grid_z = r->dispatch_info.grid_size.z;
        rec.wg_x = r->dispatch_info.workgroup_size.x;
        rec.wg_y = r->dispatch_info.workgroup_size.y;
        rec.wg_z = r->dispatch_info.workgroup_size.z;
        emit(&rec, sizeof(rec));
        break;
      }
      case ROCPROFILER_BUFFER_TRACING_MEMORY_COPY: {
        auto* r = static_cast<rocprofiler_buffer_tracing_memory_copy_record_t*>(
            header->payload);
        sgt::CopyRec rec{};
        rec.h = {sgt::REC_COPY, sizeof(sgt::CopyRec), 0};
        rec.start_ns = r->start_timestamp;
        rec.end_ns = r->end_timestamp;
        rec.corr_id = r->correlation_id.internal;
        rec.tid = static_cast<uint32_t>(r->thread_id);
        rec.op = r->operation;
        rec.src_device = agent_device(r->src_agent_id);
        rec.dst_device = agent_device(r->dst_agent_id);
        rec.bytes = r->bytes;
        emit(&rec, sizeof(rec));
        break;
      }
      case ROCPROFILER_BUFFER_TRACING_HIP_RUNTIME_API: {
        auto* r = static_cast<rocprofiler_buffer_tracing_hip_api_record_t*>(
            header->payload);
        sgt::ApiRec rec{};
        rec.h = {sgt::REC_HIPAPI, sizeof(sgt::ApiRec), 0};
        rec.start_ns = r->start_timestamp;
        rec.end_ns = r->end_timestamp;
        rec.corr_id = r->correlation_id.internal;
        rec.tid = static_cast<uint32_t>(r->thread_id);
        rec.op = r->operation;
        emit(&rec, sizeof(rec));
        break;
      }
      case ROCPROFILER_BUFFER_TRACING_KFD_EVENT_PAGE_MIGRATE: {
        auto* r = static_cast<
            rocprofiler_buffer_tracing_kfd_event_page_migrate_record_t*>(
            header->payload);
        sgt::KfdRec rec{};
        rec.h = {sgt::REC_KFD, sizeof(sgt::KfdRec), 0};
        rec.timestamp = r->timestamp;
        rec.op_class = sgt::KFD_PAGE_MIGRATE;
        rec.operation = r->operation;
        rec.pid = r->pid;
        rec.device = agent_device(r->dst_agent);
        rec.src_device = agent_device(r->src_agent);
        rec.addr_start = r->start_address.value;
        rec.addr_end = r->end_address.value;
        rec.error_code = r->error_code;
        emit(&rec, sizeof(rec));
        break;
      }
      case ROCPROFILER_BUFFER_TRACING_KFD_EVENT_PAGE_FAULT: {
        auto* r = static_cast<
            rocprofiler_buffer_tracing_kfd_event_page_fault_record_t*>(
            header->payload);
        sgt::KfdRec rec{};
        rec.h = {sgt::REC_KFD, sizeof(sgt::KfdRec), 0};
        rec.timestamp = r->timestamp;
        rec.op_class = sgt::KFD_PAGE_FAULT;
        rec.operation = r->operation;
        rec.pid = r->pid;
        rec.device = agent_device(r->agent_id);
        rec.src_device = -1;
        rec.addr_start = r->address.value;
        rec.addr_end = r->address.value;
        emit(&rec, sizeof(rec));
        break;
      }
      case ROCPROFILER_BUFFER_TRACING_MEMORY_ALLOCATION: {
        auto* r =
            static_cast<rocprofiler_buffer_tracing_memory_allocation_record_t*>(
                header->payload);
        sgt::AllocRec rec{};
        rec.h = {sgt::REC_ALLOC, sizeof(sgt::AllocRec), 0};
        rec.start_ns = r->start_timestamp;
        rec.end_ns = r->end_timestamp;
        rec.corr_id = r->correlation_id.internal;
        rec.tid = static_cast<uint32_t>(r->thread_id);
        rec.op = r->operation;
        rec.device = agent_device(r->agent_id);
        rec.address = r->address.value;
        rec.bytes = r->allocation_size;
        emit(&rec, sizeof(rec));
        break;
      }
      default:
        break;
    }
  }
  if (!chunk.empty()) {
    write_raw(chunk.data(), chunk.size());
    g_n_records.fetch_add(num_headers, std::memory_order_relaxed);
  }
}

// GPU program-counter samples (SOFA_PC_SAMPLING=1; experimental SDK API,
// host-trap method): each sample -> PcSampleRec keyed by the dispatch's
// correlation id, so preprocess attributes hotspots per kernel.  A feature
// the reference never had (nvprof exposed no PC sampling to SOFA).
void pc_buffer_callback(rocprofiler_context_id_t, rocprofiler_buffer_id_t,
                        rocprofiler_record_header_t** headers,
                        size_t num_headers, void*, uint64_t drop_count) {
  if (drop_count > 0) {
    sgt::DropRec d{};
    d.h = {sgt::REC_DROP, sizeof(sgt::DropRec), 0};
    d.dropped = drop_count;
    write_raw(&d, sizeof(d));
  }
  std::vector<char> chunk;
  chunk.reserve(num_headers * sizeof(sgt::PcSampleRec));
  size_t n = 0;
  for (size_t i = 0; i < num_headers; ++i) {
    auto* header = headers[i];
    if (header->category != ROCPROFILER_BUFFER_CATEGORY_PC_SAMPLING) continue;
    sgt::PcSampleRec rec{};
    rec.h = {sgt::REC_PCSAMPLE, sizeof(sgt::PcSampleRec), 0};
    if (header->kind == ROCPROFILER_PC_SAMPLING_RECORD_HOST_TRAP_V0_SAMPLE) {
      auto* r = static_cast<rocprofiler_pc_sampling_record_host_trap_v0_t*>(
          header->payload);
      rec.timestamp = r->timestamp;
      rec.corr_id = r->correlation_id.internal;
      rec.code_object_id = r->pc.code_object_id;
      rec.offset = r->pc.code_object_offset;
      rec.exec_mask = r->exec_mask;
      rec.dispatch_id = r->dispatch_id;
      rec.wave_in_group = r->wave_in_group;
    } else if (header->kind ==
               ROCPROFILER_PC_SAMPLING_RECORD_STOCHASTIC_V0_SAMPLE) {
      auto* r = static_cast<rocprofiler_pc_sampling_record_stochastic_v0_t*>(
          header->payload);
      rec.timestamp = r->timestamp;
      rec.corr_id = r->correlation_id.internal;
      rec.code_object_id = r->pc.code_object_id;
      rec.offset = r->pc.code_object_offset;
      rec.exec_mask = r->exec_mask;
      rec.dispatch_id = r->dispatch_id;
      rec.wave_in_group = r->wave_in_group;
    } else {
      continue;
    }
    rec.device = 0;
    chunk.insert(chunk.end(), (const char*) &rec,
                 (const char*) &rec + sizeof(rec));
    ++n;
  }
  if (!chunk.empty()) {
    write_raw(chunk.data(), chunk.size());
    g_n_records.fetch_add(n, std::memory_order_relaxed);
  }
}

struct PcCfgPick {
  bool found = false;
  rocprofiler_pc_sampling_method_t method{};
  rocprofiler_pc_sampling_unit_t unit{};
  uint64_t interval = 0;
};

void configure_pc_sampling() {
  size_t buffer_bytes = 8 << 20;
  if (rocprofiler_create_buffer(g_ctx, buffer_bytes, buffer_bytes / 2,
                                ROCPROFILER_BUFFER_POLICY_LOSSLESS,
                                pc_buffer_callback, nullptr,
                                &g_pc_buffer) != ROCPROFILER_STATUS_SUCCESS)
    return;
  uint64_t req_us = 200;  // default ~5 kHz host-trap
  if (const char* v = getenv("SOFA_PC_SAMPLING_INTERVAL_US"); v && *v)
    req_us = strtoull(v, nullptr, 10);
  for (auto agent : g_gpu_agents) {
    PcCfgPick pick;
    rocprofiler_query_pc_sampling_agent_configurations(
        agent,
        [](const rocprofiler_pc_sampling_configuration_t* cfgs, size_t n,
           void* ud) -> rocprofiler_status_t {
          auto* p = static_cast<PcCfgPick*>(ud);
          for (size_t i = 0; i < n; ++i) {
            bool host_trap =
                cfgs[i].method == ROCPROFILER_PC_SAMPLING_METHOD_HOST_TRAP;
            bool stochastic =
                cfgs[i].method == ROCPROFILER_PC_SAMPLING_METHOD_STOCHASTIC;
            if (!host_trap && !stochastic) continue;
            // prefer host-trap (time-based, simple); stochastic is the
            // MI300+/gfx950 hardware sampler
            if (p->found && p->method == ROCPROFILER_PC_SAMPLING_METHOD_HOST_TRAP)
              continue;
            p->found = true;
            p->method = cfgs[i].method;
            p->unit = cfgs[i].unit;
            p->interval = cfgs[i].min_interval;  // caller raises to request
          }
          return ROCPROFILER_STATUS_SUCCESS;
        },
        &pick);
    if (!pick.found) {
      fprintf(stderr,
              "[sofatracer] pc sampling unavailable (agent %lx): driver offers no "
              "host-trap or stochastic configuration\n",
              (unsigned long) agent.handle);
      continue;
    }
    uint64_t interval = req_us * 1000;  // ns when unit == TIME
    if (pick.unit != ROCPROFILER_PC_SAMPLING_UNIT_TIME) {
      // cycles/instructions unit (stochastic): use a power of two around
      // ~1M cycles (~0.5 ms at 2 GHz) unless the minimum is higher
      interval = 1u << 20;
    }
    if (interval < pick.interval) interval = pick.interval;
    auto st = rocprofiler_configure_pc_sampling_service(
        g_ctx, agent, pick.method, pick.unit, interval, g_pc_buffer, 0);
    if (st == ROCPROFILER_STATUS_SUCCESS) {
      g_pc_active = true;
      fprintf(stderr,
              "[sofatracer] pc sampling active (agent %lx, interval %lu)\n",
              (unsigned long) agent.handle, (unsigned long) interval);
    } else {
      fprintf(stderr, "[sofatracer] pc sampling unavailable (agent %lx): %s\n",
              (unsigned long) agent.handle,
              rocprofiler_get_status_string(st));
    }
  }
}

void write_agents() {
  rocprofiler_query_available_agents(
      ROCPROFILER_AGENT_INFO_VERSION_0,
      [](rocprofiler_agent_version_t, const void** agents, size_t num_agents,
         void*) -> rocprofiler_status_t {
        for (size_t i = 0; i < num_agents; ++i) {
          const auto* a =
              static_cast<const rocprofiler_agent_v0_t*>(agents[i]);
          int32_t dev = (a->type == ROCPROFILER_AGENT_TYPE_GPU)
                            ? a->logical_node_type_id
                            : -1;
          agent_device_map()[a->id.handle] = dev;
          if (dev >= 0) g_gpu_agents.push_back(a->id);
          sgt::AgentRec rec{};
          rec.h = {sgt::REC_AGENT, sizeof(sgt::AgentRec), 0};
          rec.agent_handle = a->id.handle;
          rec.device = dev;
          rec.type = a->type;
          rec.node_id = a->node_id;
          rec.wave_front_size = a->wave_front_size;
          rec.cu_count = a->cu_count;
          rec.num_xcc = a->num_xcc;
          if (a->name) {
            strncpy(rec.name, a->name, sizeof(rec.name) - 1);
          }
          write_raw(&rec, sizeof(rec));
        }
        return ROCPROFILER_STATUS_SUCCESS;
      },
      sizeof(rocprofiler_agent_v0_t), nullptr);
}

void write_opnames() {
  // dump operation-name tables for the kinds we record so preprocess never
  // needs the SDK at analysis time
  for (auto kind : {ROCPROFILER_BUFFER_TRACING_HIP_RUNTIME_API,
                    ROCPROFILER_BUFFER_TRACING_MEMORY_COPY,
                    ROCPROFILER_BUFFER_TRACING_MEMORY_ALLOCATION}) {
    rocprofiler_iterate_buffer_tracing_kind_operations(
        kind,
        [](rocprofiler_buffer_tracing_kind_t k, rocprofiler_tracing_operation_t op,
           void*) -> int {
          const char* name = nullptr;
          uint64_t len = 0;
          if (rocprofiler_query_buffer_tracing_kind_operation_name(
                  k, op, &name, &len) == ROCPROFILER_STATUS_SUCCESS &&
              name) {
            write_opname_rec(k, op, name);
          }
          return 0;  // 0 = continue iterating (nonzero stops)
        },
        nullptr);
  }
  // RCCL records come from CALLBACK tracing: its op-name table lives under
  // the callback kind, not the buffer kind
  rocprofiler_iterate_callback_tracing_kind_operations(
      ROCPROFILER_CALLBACK_TRACING_RCCL_API,
      [](rocprofiler_callback_tracing_kind_t k, rocprofiler_tracing_operation_t op,
         void*) -> int {
        const char* name = nullptr;
        uint64_t len = 0;
        if (rocprofiler_query_callback_tracing_kind_operation_name(
                k, op, &name, &len) == ROCPROFILER_STATUS_SUCCESS &&
            name) {
          write_opname_rec(1000 + k, op, name);  // offset: callback-kind space
        }
        return 0;  // 0 = continue iterating
      },
      nullptr);
}

int tool_init(rocprofiler_client_finalize_t, void*) {
  g_null_sink = env_flag("SOFA_NULL_SINK", false);
  const char* logdir = getenv("SOFA_LOGDIR");
  if (!logdir || !*logdir) logdir = ".";
  char path[4096];
  snprintf(path, sizeof(path), "%s/gputrace_%d.sgt", logdir, getpid());
  g_out = fopen(path, "wb");
  if (!g_out) {
    fprintf(stderr, "[sofatracer] cannot open %s\n", path);
    return -1;
  }

  sgt::FileHeader hdr{};
  hdr.magic = sgt::kMagic;
  hdr.version = sgt::kVersion;
  hdr.pid = static_cast<uint32_t>(getpid());
  rocprofiler_timestamp_t ts = 0;
  rocprofiler_get_timestamp(&ts);
  hdr.realtime_ns = host_ns(CLOCK_REALTIME);
  hdr.monotonic_raw_ns = host_ns(CLOCK_MONOTONIC_RAW);
  hdr.rocp_ns = ts;
  fwrite(&hdr, sizeof(hdr), 1, g_out);

  write_agents();
  write_opnames();
  write_clock_rec();

  if (rocprofiler_create_context(&g_ctx) != ROCPROFILER_STATUS_SUCCESS)
    return -1;

  auto code_object_ops = std::vector<rocprofiler_tracing_operation_t>{
      ROCPROFILER_CODE_OBJECT_DEVICE_KERNEL_SYMBOL_REGISTER};
  rocprofiler_configure_callback_tracing_service(
      g_ctx, ROCPROFILER_CALLBACK_TRACING_CODE_OBJECT, code_object_ops.data(),
      code_object_ops.size(), code_object_callback, nullptr);

  size_t buffer_mb = 64;
  if (const char* v = getenv("SOFA_GPU_BUFFER_MB"); v && *v)
    buffer_mb = strtoull(v, nullptr, 10);
  size_t buffer_bytes = buffer_mb << 20;
  if (rocprofiler_create_buffer(g_ctx, buffer_bytes, buffer_bytes / 2,
                                ROCPROFILER_BUFFER_POLICY_LOSSLESS,
                                buffer_callback, nullptr,
                                &g_buffer) != ROCPROFILER_STATUS_SUCCESS)
    return -1;

  // SOFA_TRACE_DISPATCH=0: RCCL/marker-only mode — no kernel/copy buffer
  // services, so the SDK installs no per-dispatch interception and this
  // library composes with libsofahsalite (which owns the dispatch timeline
  // at a fraction of the cost; see hsalite/hsalite.cc)
  bool trace_dispatch = env_flag("SOFA_TRACE_DISPATCH", true);
  if (trace_dispatch) {
    rocprofiler_configure_buffer_tracing_service(
        g_ctx, ROCPROFILER_BUFFER_TRACING_KERNEL_DISPATCH, nullptr, 0, g_buffer);
  }
  // Copies are low-rate (tens per step), so SDK copy tracing is ~free and
  // stays on in the lite hybrid (SOFA_TRACE_COPY=1 SOFA_TRACE_DISPATCH=0):
  // measured on MI355X, ROCclr's D2H/H2D SDMA path does not go through the
  // public hsa_amd_memory_async_copy* entries hsalite wraps.
  if (env_flag("SOFA_TRACE_COPY", trace_dispatch)) {
    rocprofiler_configure_buffer_tracing_service(
        g_ctx, ROCPROFILER_BUFFER_TRACING_MEMORY_COPY, nullptr, 0, g_buffer);
  }
  if (trace_dispatch && env_flag("SOFA_TRACE_HIP_API", true)) {
    // SOFA_HIP_API_OPS: "all" = every HIP call (expensive at launch rates),
    // default = launches/copies/syncs/allocs only (the calls a timeline
    // reader actually drills into; cuts record volume massively)
    const char* ops_sel = getenv("SOFA_HIP_API_OPS");
    bool all_ops = ops_sel && strcmp(ops_sel, "all") == 0;
    if (all_ops) {
      rocprofiler_configure_buffer_tracing_service(
          g_ctx, ROCPROFILER_BUFFER_TRACING_HIP_RUNTIME_API, nullptr, 0,
          g_buffer);
    } else {
      static std::vector<rocprofiler_tracing_operation_t> ops;
      rocprofiler_iterate_buffer_tracing_kind_operations(
          ROCPROFILER_BUFFER_TRACING_HIP_RUNTIME_API,
          [](rocprofiler_buffer_tracing_kind_t k,
             rocprofiler_tracing_operation_t op, void*) -> int {
            const char* name = nullptr;
            uint64_t len = 0;
            if (rocprofiler_query_buffer_tracing_kind_operation_name(
                    k, op, &name, &len) == ROCPROFILER_STATUS_SUCCESS &&
                name) {
              std::string_view n{name};
              if (n.find("Launch") != std::string_view::npos ||
                  n.find("Memcpy") != std::string_view::npos ||
                  n.find("Memset") != std::string_view::npos ||
                  n.find("Synchronize") != std::string_view::npos ||
                  n.find("Malloc") != std::string_view::npos ||
                  n.find("hipFree") != std::string_view::npos ||
                  n.find("GraphLaunch") != std::string_view::npos ||
                  n.find("EventRecord") != std::string_view::npos ||
                  n.find("StreamWaitEvent") != std::string_view::npos) {
                ops.push_back(op);
              }
            }
            return 0;
          },
          nullptr);
      rocprofiler_configure_buffer_tracing_service(
          g_ctx, ROCPROFILER_BUFFER_TRACING_HIP_RUNTIME_API, ops.data(),
          ops.size(), g_buffer);
    }
  }
  if (env_flag("SOFA_TRACE_ALLOC", false)) {
    rocprofiler_configure_buffer_tracing_service(
        g_ctx, ROCPROFILER_BUFFER_TRACING_MEMORY_ALLOCATION, nullptr, 0,
        g_buffer);
  }
  if (env_flag("SOFA_TRACE_KFD", false)) {
    // page-migrate/fault events (SVM memory pressure diagnosis)
    rocprofiler_configure_buffer_tracing_service(
        g_ctx, ROCPROFILER_BUFFER_TRACING_KFD_EVENT_PAGE_MIGRATE, nullptr, 0,
        g_buffer);
    rocprofiler_configure_buffer_tracing_service(
        g_ctx, ROCPROFILER_BUFFER_TRACING_KFD_EVENT_PAGE_FAULT, nullptr, 0,
        g_buffer);
  }
  if (env_flag("SOFA_TRACE_RCCL", true)) {
    rocprofiler_configure_callback_tracing_service(
        g_ctx, ROCPROFILER_CALLBACK_TRACING_RCCL_API, nullptr, 0,
        rccl_callback, nullptr);
  }
  rocprofiler_configure_callback_tracing_service(
      g_ctx, ROCPROFILER_CALLBACK_TRACING_MARKER_CORE_API, nullptr, 0,
      marker_callback, nullptr);
  if (env_flag("SOFA_PC_SAMPLING", false)) configure_pc_sampling();

  auto cb_thread = rocprofiler_callback_thread_t{};
  if (rocprofiler_create_callback_thread(&cb_thread) ==
      ROCPROFILER_STATUS_SUCCESS)
    rocprofiler_assign_callback_thread(g_buffer, cb_thread);

  int valid = 0;
  rocprofiler_context_is_valid(g_ctx, &valid);
  if (valid == 0) return -1;
  // SOFA_DEFER_START=1: stay disarmed until sofa_tracer_start() (bench.py's
  // in-process overhead A/B measurement)
  if (!env_flag("SOFA_DEFER_START", false)) rocprofiler_start_context(g_ctx);
  return 0;
}

void tool_fini(void*) {
  rocprofiler_flush_buffer(g_buffer);
  write_clock_rec();
  std::lock_guard<std::mutex> lk(g_mutex);
  if (g_out) {
    fclose(g_out);
    g_out = nullptr;
  }
}

}  // namespace

// ---- runtime control API (dlopen'd by bench.py / sofa_amd via ctypes) ----

extern "C" int sofa_tracer_start() {
  if (g_ctx.handle == 0) return -1;
  return rocprofiler_start_context(g_ctx) == ROCPROFILER_STATUS_SUCCESS ? 0 : -1;
}

extern "C" int sofa_tracer_stop() {
  if (g_ctx.handle == 0) return -1;
  auto st = rocprofiler_stop_context(g_ctx);
  rocprofiler_flush_buffer(g_buffer);
  return st == ROCPROFILER_STATUS_SUCCESS ? 0 : -1;
}

extern "C" unsigned long long sofa_tracer_event_count() {
  rocprofiler_flush_buffer(g_buffer);
  return g_n_records.load(std::memory_order_relaxed);
}

extern "C" int sofa_tracer_active() { return g_out != nullptr; }

extern "C" int sofa_tracer_pc_sampling_active() { return g_pc_active ? 1 : 0; }

extern "C" rocprofiler_tool_configure_result_t*
rocprofiler_configure(uint32_t version, const char* runtime_version,
                      uint32_t priority, rocprofiler_client_id_t* id) {
  id->name = "sofa_amd-tracer";
  g_client_id = id;
  (void) version;
  (void) runtime_version;
  (void) priority;
  static auto cfg = rocprofiler_tool_configure_result_t{
      sizeof(rocprofiler_tool_configure_result_t), &tool_init, &tool_fini,
      nullptr};
  return &cfg;
}
