"""Configuration for sofa_amd.

Replaces the reference's mutable class-attribute bag ``SOFA_Config``
(cyliustack/sofa bin/sofa_config.py:10-74) with a typed dataclass; the unified
13-column trace schema columns are kept verbatim so downstream CSVs stay
compatible with the reference's layout (bin/sofa_config.py:49-62).
"""

from __future__ import annotations

import dataclasses
import os
from dataclasses import dataclass, field
from typing import List, Optional

# The unified trace schema (reference bin/sofa_config.py:49-62).  Semantics are
# source-dependent: for CPU samples `event` = log10(ip) and `duration` =
# period/MHz; for GPU rows `copyKind` encodes the copy class (0 kernel, 1 H2D,
# 2 D2H, 8 D2D, 10 P2P, 16 RCCL collective), kernel rows carry the workgroup
# count in `payload` and LDS bytes in `pkt_src` (chip-underfill analysis);
# for network rows pkt_src/pkt_dst are IPv4 addresses packed base-1000.
TRACE_COLUMNS = [
    "timestamp",
    "event",
    "duration",
    "deviceId",
    "copyKind",
    "payload",
    "bandwidth",
    "pkt_src",
    "pkt_dst",
    "pid",
    "tid",
    "name",
    "category",
]

# copyKind codes (reference bin/sofa_common.py:20-21 cktable); 16 is new here:
# an RCCL collective op attributed to xGMI links.
CKTABLE = {
    -1: "TOTAL",
    0: "KERNEL",
    1: "H2D",
    2: "D2H",
    8: "D2D",
    10: "P2P",
    16: "RCCL",
}


@dataclass
class Filter:
    """Keyword -> display color filter (reference bin/sofa_config.py:1-7)."""

    keyword: str
    color: str


@dataclass
class SofaConfig:
    """All knobs for record/preprocess/analyze/viz.

    Field defaults mirror the reference's observable defaults
    (bin/sofa_config.py:10-74) where a matching concept exists.
    """

    logdir: str = "sofalog"
    command: str = ""
    verbose: bool = False

    # --- record ---
    cpu_sample_rate: int = 99          # Hz, reference perf record -F 99
    sys_mon_rate: int = 10             # Hz, /proc + rocm-smi pollers
    profile_all_cpus: bool = False     # system-wide CPU sampling (-a)
    perf_events: str = "cpu-clock"     # no hw PMU assumed; sw cpu-clock sampler
    enable_tcpdump: bool = False       # AF_PACKET sniffer
    enable_strace: bool = False
    enable_pystacks: bool = False
    enable_callchain: bool = False   # -g native call stacks -> flamegraph
    enable_vmstat: bool = True
    enable_diskstat: bool = True
    enable_netstat: bool = True
    enable_gpu: bool = True            # rocprofiler-sdk collector
    # HIP runtime API spans: default ON with the FILTERED op set (launches/
    # copies/syncs/allocs — measured ~0 extra overhead at 2x events/s);
    # full API tracing (--hip_api_full) costs ~45% on launch-dense steps
    # (profiles/overhead_matrix_r01.md)
    # "lite" (default) = HSA-level dispatch/copy tracer, 3.4% measured
    # overhead on ResNet-50 bs=64 (profiles/overhead_decomp_r02.md), RCCL
    # args via debug-log channel; "sdk" = rocprofiler-sdk collector (full
    # fidelity: HIP API spans, RCCL args with corr ids, KFD events, allocs)
    # at ~13% overhead.  The two cannot share a process (ROCr skips
    # HSA_TOOLS_LIB tools when rocprofiler is registered - measured).
    gpu_tracer: str = "lite"
    enable_gpu_hip_api: bool = True
    hip_api_full: bool = False
    enable_rccl_trace: bool = True     # RCCL API tracing via collector
    rccl_shim: bool = False            # LD_PRELOAD interposer (fallback path)
    enable_kfd_trace: bool = False     # page-migrate/fault events
    pc_sampling: bool = False          # GPU PC sampling (sdk collector, experimental)
    gpu_sample: int = 1                # lite: time every Nth dispatch (overhead ~1/N)
    gpu_ring_buffer_mb: int = 64       # collector buffer size per process
    blkdev: str = ""                   # block device for blktrace-like stats
    enable_blkio: bool = False         # tracefs block_rq_issue/complete per-IO tracing
    docker_image: str = ""             # profile the command inside this container image
    nvsmi_interval_ms: int = 100       # GPU telemetry poll period

    # --- preprocess ---
    cpu_time_offset_ms: int = 0
    plot_ratio: int = 1                # downsample ratio for viz series
    strace_min_time: float = 1e-4
    cpu_filters: List[Filter] = field(default_factory=list)
    gpu_filters: List[Filter] = field(default_factory=list)
    net_filters: List[Filter] = field(default_factory=list)
    diskstat_filters: List[Filter] = field(default_factory=list)
    enable_swarms: bool = False
    num_swarms: int = 10

    # --- analyze ---
    num_iterations: int = 20
    enable_aisi: bool = False
    aisi_via_strace: bool = False
    is_idle_threshold: int = 10        # % util below which a window is idle
    spotlight_gpu: bool = False
    profile_region: Optional[str] = None  # "begin,end" seconds into the run
    roi_begin: float = 0.0
    roi_end: float = 0.0
    potato_server: str = ""
    cluster_ip: str = ""
    base_logdir: str = "sofalog-base"
    match_logdir: str = "sofalog-match"
    skip_preprocess: bool = False

    # --- viz ---
    viz_port: int = 8000
    viz_host: str = "127.0.0.1"

    # populated at runtime
    time_base: float = 0.0

    def __post_init__(self) -> None:
        if not self.logdir.endswith("/"):
            self.logdir += "/"
        # default GPU filters (reference bin/sofa:273-286): copies + fw/bw +
        # allreduce kernels; RCCL device kernels on ROCm carry "rccl"/"Ccl".
        if not self.gpu_filters:
            self.gpu_filters = [
                Filter("CopyHostToDevice", "Red"),
                Filter("CopyDeviceToHost", "Peru"),
                Filter("CopyPeerToPeer", "Purple"),
                Filter("rccl", "indigo"),
            ]

    @property
    def logdir_path(self) -> str:
        return self.logdir

    def to_dict(self) -> dict:
        d = dataclasses.asdict(self)
        return d


def ensure_logdir(cfg: SofaConfig) -> str:
    os.makedirs(cfg.logdir, exist_ok=True)
    return cfg.logdir
