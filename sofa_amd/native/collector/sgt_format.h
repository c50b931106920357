// SGT ("sofa gpu trace") binary format — shared between the rocprofiler-sdk
// collector library (writer) and sofa_amd.preprocess.gpu (reader; a mirrored
// numpy structured dtype lives in sofa_amd/preprocess/sgt.py — keep in sync).
//
// All records are little-endian, 8-byte aligned, length-prefixed.

#pragma once

#include <cstdint>

namespace sgt {

constexpr uint32_t kMagic = 0x31544753;  // "SGT1"
constexpr uint32_t kVersion = 1;

struct FileHeader {
  uint32_t magic;
  uint32_t version;
  uint32_t pid;
  uint32_t reserved0;
  // clock correlation sample at init: one instant in three clocks
  uint64_t realtime_ns;
  uint64_t monotonic_raw_ns;
  uint64_t rocp_ns;  // rocprofiler_get_timestamp at the same instant
  uint64_t reserved[3];
};

enum RecType : uint16_t {
  REC_KERNEL = 1,       // KernelRec
  REC_COPY = 2,         // CopyRec
  REC_HIPAPI = 3,       // ApiRec
  REC_RCCL = 4,         // RcclRec
  REC_KERNEL_NAME = 5,  // NameRec (kernel_id -> mangled name)
  REC_OPNAME = 6,       // OpNameRec (kind,op -> name)
  REC_AGENT = 7,        // AgentRec
  REC_CLOCK = 8,        // ClockRec (emitted at init + fini for drift)
  REC_ALLOC = 9,        // AllocRec
  REC_DROP = 10,        // DropRec (buffer drops, should be 0 for lossless)
  REC_MARKER = 11,      // NameRec-shaped: roctx range/instant (id -> message)
  REC_KFD = 12,         // KfdRec: page migrate/fault/queue events
  REC_PCSAMPLE = 13,    // PcSampleRec: GPU program-counter sample
};

// KFD event classes (KfdRec.op_class)
enum KfdClass : uint32_t {
  KFD_PAGE_MIGRATE = 1,
  KFD_PAGE_FAULT = 2,
  KFD_QUEUE_EVT = 3,
  KFD_UNMAP = 4,
};

struct RecHeader {
  uint16_t type;
  uint16_t size;  // total record bytes including this header (8-byte mult)
  uint32_t pad;
};

struct KernelRec {
  RecHeader h;           // REC_KERNEL
  uint64_t start_ns;     // rocprofiler clock
  uint64_t end_ns;
  uint64_t corr_id;
  uint32_t tid;
  uint32_t device;       // logical GPU index (HIP_VISIBLE_DEVICES order)
  uint64_t queue_id;
  uint64_t kernel_id;    // join with REC_KERNEL_NAME
  uint32_t private_segment_size;
  uint32_t group_segment_size;  // LDS bytes
  uint32_t grid_x, grid_y, grid_z;
  uint32_t wg_x, wg_y, wg_z;
  uint32_t pad2;
};  // 96 bytes

struct CopyRec {
  RecHeader h;        // REC_COPY
  uint64_t start_ns;
  uint64_t end_ns;
  uint64_t corr_id;
  uint32_t tid;
  uint32_t op;        // rocprofiler_memory_copy_operation_t
  int32_t src_device; // logical GPU index or -1 for host
  int32_t dst_device;
  uint64_t bytes;
};  // 56 bytes

struct ApiRec {
  RecHeader h;        // REC_HIPAPI
  uint64_t start_ns;
  uint64_t end_ns;
  uint64_t corr_id;
  uint32_t tid;
  uint32_t op;        // rocprofiler_hip_runtime_api_id_t
};  // 40 bytes

struct RcclRec {
  RecHeader h;        // REC_RCCL
  uint64_t start_ns;
  uint64_t end_ns;
  uint64_t corr_id;
  uint32_t tid;
  uint32_t op;        // rocprofiler_rccl_api_id_t
  uint64_t count;     // element count (0 when not applicable)
  uint32_t datatype;  // ncclDataType_t
  uint32_t elem_size; // bytes per element (precomputed)
  int32_t peer_or_root;  // peer rank (send/recv), root (bcast/reduce), else -1
  uint32_t device;    // current HIP device at call time
  uint64_t comm;      // ncclComm_t pointer value (communicator identity)
  uint64_t stream;    // hipStream_t pointer value
};  // 80 bytes

struct NameRec {  // REC_KERNEL_NAME / REC_MARKER: header + id + chars
  RecHeader h;
  uint64_t id;
  // char name[]; NUL-terminated, padded to 8 bytes
};

struct OpNameRec {  // REC_OPNAME: header + kind + op + chars
  RecHeader h;
  uint32_t kind;
  uint32_t op;
  // char name[];
};

struct AgentRec {
  RecHeader h;          // REC_AGENT
  uint64_t agent_handle;
  int32_t device;       // logical_node_type_id for GPUs, -1 for CPU agents
  int32_t type;         // rocprofiler_agent_type_t (1=CPU, 2=GPU)
  uint32_t node_id;
  uint32_t wave_front_size;
  uint32_t cu_count;
  uint32_t num_xcc;
  char name[64];        // gfx name, NUL-terminated
};  // 104 bytes

struct ClockRec {
  RecHeader h;  // REC_CLOCK
  uint64_t realtime_ns;
  uint64_t monotonic_raw_ns;
  uint64_t rocp_ns;
};  // 32 bytes

struct AllocRec {
  RecHeader h;     // REC_ALLOC
  uint64_t start_ns;
  uint64_t end_ns;
  uint64_t corr_id;
  uint32_t tid;
  uint32_t op;     // rocprofiler_memory_allocation_operation_t
  int32_t device;
  uint32_t pad2;
  uint64_t address;
  uint64_t bytes;
};  // 64 bytes

struct DropRec {
  RecHeader h;  // REC_DROP
  uint64_t dropped;
};

struct KfdRec {
  RecHeader h;        // REC_KFD
  uint64_t timestamp; // KFD-reported ns (rocprofiler clock domain)
  uint32_t op_class;  // KfdClass
  uint32_t operation; // start/end/etc per class
  uint32_t pid;
  int32_t device;     // logical GPU (or -1)
  uint64_t addr_start;
  uint64_t addr_end;
  int32_t src_device; // migrate: source agent's logical id
  int32_t error_code;
};  // 56 bytes

struct PcSampleRec {
  RecHeader h;              // REC_PCSAMPLE
  uint64_t timestamp;       // rocprofiler clock ns
  uint64_t corr_id;         // joins the owning kernel dispatch
  uint64_t code_object_id;
  uint64_t offset;          // PC offset within the code object
  uint64_t exec_mask;       // active SIMD lanes at sample time
  uint64_t dispatch_id;
  uint32_t wave_in_group;
  uint32_t device;          // logical GPU index
};  // 64 bytes

}  // namespace sgt
