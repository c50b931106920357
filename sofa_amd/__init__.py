"""sofa_amd — MI355X-native whole-system performance profiler.

A from-scratch rebuild of the capabilities of cyliustack/sofa
designed for ROCm/CDNA4 systems:

* ``sofa record "<cmd>"`` wraps an arbitrary command and concurrently records
  CPU samples (perf_event_open-based native sampler), GPU kernel/copy/HIP-API/RCCL
  activity (rocprofiler-sdk tool library -> binary ring trace), GPU telemetry and
  xGMI topology (librocm_smi64 via ctypes), network packets (AF_PACKET sniffer)
  and bandwidth, disk and VM statistics.
* ``sofa preprocess`` merges every stream onto one clock-synchronized timeline in
  the unified 13-column trace schema and emits CSVs + report.js.
* ``sofa analyze`` computes per-subsystem profiles, communication matrices with
  per-xGMI-link RCCL attribution, concurrency breakdown, iteration detection
  (suffix tree), swarm clustering, ring-order hints and a feature vector.
* ``sofa viz`` serves the sofaboard HTML dashboard.

Reference behavior map: cyliustack/sofa bin/sofa{,_record,_preprocess,_analyze}.py
(layout documented in SURVEY.md §1-§3). The implementation here is new and
MI355X-first: no nvprof/CUPTI paths, no CUDA shims.
"""

__version__ = "0.1.0"
