"""sofa record — orchestration.

Behavioral parity with reference bin/sofa_record.py:150-523 (preflight,
prologue clock sync, background monitors, target launch, epilogue), rebuilt
for ROCm:

* GPU activity: rocprofiler-sdk collector library injected via
  ROCP_TOOL_LIBRARIES (replaces the system-wide nvprof daemon,
  bin/sofa_record.py:217-221).
* CPU samples: native sofa-cpusampler on perf_event_open (replaces
  `perf record -F 99`, :339-354).
* Clock sync: every collector stamps CLOCK_MONOTONIC_RAW + CLOCK_REALTIME
  directly (replaces the sofa_perf_timebase + cuhello pairing,
  :236-242; see native/timebase/timebase.cc).
* Telemetry: one SysMonitor thread + librocm_smi64 ctypes
  (replaces ~10 monitor subprocesses, :249-312).
"""

from __future__ import annotations

import json
import os
import shutil
import signal
import subprocess
import sys
import time

from ..config import SofaConfig, ensure_logdir
from .. import printing as p
from .pollers import SysMonitor, dump_xgmi_topology

RAW_FILES = [
    "cpusamples.scs",
    "timebase.json",
    "sofa_time.txt",
    "cpuinfo.txt",
    "mpstat.txt",
    "diskstat.txt",
    "netstat.txt",
    "vmstat.txt",
    "gpusmi.txt",
    "xgmi_topo.txt",
    "kallsyms",
    "misc.txt",
    "pktcap.bin",
    "strace.sst",
    "sofa_pids.txt",
    "sofa.err",
]

DERIVED_FILES = [
    "cputrace.csv",
    "gputrace.csv",
    "rccltrace.csv",
    "comm.csv",
    "mpstat.csv",
    "usr_sys.csv",
    "diskstat.csv",
    "diskstat_vector.csv",
    "diskstat_vector_ui.csv",
    "vmstat.csv",
    "netstat.csv",
    "netbandwidth.csv",
    "nettrace.csv",
    "gpusmi_trace.csv",
    "hip_api_trace.csv",
    "report.js",
    "performance.csv",
    "netrank.csv",
    "auto_caption.csv",
    "swarms_report.txt",
    "iteration_timeline.txt",
    "features.csv",
    "xlink_traffic.csv",
    "kfdtrace.csv",
    "markers.csv",
    "strace.csv",
    "pystacks.csv",
    "chrome_trace.json",
    "flamegraph.folded",
    "gpu_timebase.json",
    "comm_payload_matrix.csv",
    "comm_bandwidth_matrix.csv",
    "correlation.csv",
    "cluster_report.csv",
    "potato_report.html",
    "swarm_diff.csv",
]


def native_dir() -> str:
    return os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "native")


def native_bin(name: str) -> str:
    return os.path.join(native_dir(), "bin", name)


def native_lib(name: str) -> str:
    return os.path.join(native_dir(), "lib", name)


def ensure_native_built(verbose: bool = False) -> bool:
    """Build native helpers on demand (first run on a new machine)."""
    needed = [native_bin("sofa-cpusampler"), native_bin("sofa-timebase"), native_lib("libsofatracer.so")]
    if all(os.path.exists(x) for x in needed):
        return True
    try:
        from ..native.build import build_all

        build_all(verbose=verbose)
    except Exception as e:  # pragma: no cover
        p.print_warning(f"native build failed: {e}")
    return all(os.path.exists(x) for x in needed)


def sofa_clean(cfg: SofaConfig) -> None:
    """Delete raw + derived artifacts (reference bin/sofa_record.py:138-147)."""
    logdir = cfg.logdir
    if not os.path.isdir(logdir):
        return
    for name in os.listdir(logdir):
        full = os.path.join(logdir, name)
        if (
            name in RAW_FILES
            or name in DERIVED_FILES
            or name.startswith(("gputrace_", "sofa_hints", "pystacks.txt"))
            or name.endswith((".sgt", ".scs", ".sst"))
        ):
            if os.path.isdir(full):
                shutil.rmtree(full, ignore_errors=True)
            else:
                try:
                    os.remove(full)
                except OSError:
                    pass
    p.print_progress(f"cleaned {logdir}")


def _write_timebase(logdir: str) -> None:
    tb_bin = native_bin("sofa-timebase")
    out = os.path.join(logdir, "timebase.json")
    try:
        with open(out, "w") as f:
            subprocess.run([tb_bin, "3"], stdout=f, check=True, timeout=10)
    except (OSError, subprocess.SubprocessError):
        # fallback: do it in python (time.clock_gettime has every clock we need)
        with open(out, "w") as f:
            for _ in range(3):
                f.write(
                    json.dumps(
                        {
                            "realtime_ns": time.clock_gettime_ns(time.CLOCK_REALTIME),
                            "monotonic_ns": time.clock_gettime_ns(time.CLOCK_MONOTONIC),
                            "monotonic_raw_ns": time.clock_gettime_ns(time.CLOCK_MONOTONIC_RAW),
                            "boottime_ns": time.clock_gettime_ns(time.CLOCK_BOOTTIME),
                        }
                    )
                    + "\n"
                )


def build_target_env(cfg: SofaConfig) -> dict:
    env = dict(os.environ)
    if cfg.enable_gpu:
        tracer = native_lib("libsofatracer.so")
        lite = native_lib("libsofahsalite.so")
        mode = getattr(cfg, "gpu_tracer", "sdk")
        if mode == "lite" and not os.path.exists(lite):
            mode = "sdk"  # graceful fallback
        if mode == "lite" and os.environ.get("ROCP_TOOL_LIBRARIES"):
            # ROCr skips HSA_TOOLS_LIB tools when rocprofiler is registered
            # (measured; see hsalite.cc) — a pre-set ROCP_TOOL_LIBRARIES
            # (e.g. running under rocprofv3) would silently disable the lite
            # collector, so fall back to the sdk path and say so
            p.print_warning(
                "ROCP_TOOL_LIBRARIES is set in the environment; the lite "
                "collector cannot coexist with rocprofiler — using "
                "--gpu_tracer sdk for this run"
            )
            mode = "sdk"
        if mode == "lite" and os.path.exists(lite):
            # HSA-level dispatch/copy tracer (lowest overhead, 2.9% measured
            # vs the SDK's 12.8% on ResNet-50 bs=64; hsalite/hsalite.cc).
            # No rocprofiler in the process: loading ROCP_TOOL_LIBRARIES
            # alongside prevents hsalite's OnLoad from firing (measured,
            # lite_probe B), so collective args come from RCCL's own debug
            # log channel instead (preprocess/rccl_log.py).
            prev_hsa = env.get("HSA_TOOLS_LIB", "")
            env["HSA_TOOLS_LIB"] = lite + ((" " + prev_hsa) if prev_hsa else "")
            env["SOFA_LOGDIR"] = os.path.abspath(cfg.logdir)
            if getattr(cfg, "gpu_sample", 1) > 1:
                env["SOFA_LITE_SAMPLE"] = str(cfg.gpu_sample)
            if cfg.enable_rccl_trace and "NCCL_DEBUG" not in env:
                env["NCCL_DEBUG"] = "INFO"
                env["NCCL_DEBUG_SUBSYS"] = "COLL"
                env["NCCL_DEBUG_FILE"] = os.path.join(
                    os.path.abspath(cfg.logdir), "rccl_debug.%h.%p"
                )
        elif os.path.exists(tracer):
            prev = env.get("ROCP_TOOL_LIBRARIES", "")
            env["ROCP_TOOL_LIBRARIES"] = tracer + ((":" + prev) if prev else "")
            env["SOFA_LOGDIR"] = os.path.abspath(cfg.logdir)
            env["SOFA_TRACE_HIP_API"] = "1" if cfg.enable_gpu_hip_api else "0"
            if getattr(cfg, "hip_api_full", False):
                env["SOFA_HIP_API_OPS"] = "all"
            env["SOFA_TRACE_RCCL"] = "1" if cfg.enable_rccl_trace else "0"
            env["SOFA_GPU_BUFFER_MB"] = str(cfg.gpu_ring_buffer_mb)
            if cfg.enable_kfd_trace:
                env["SOFA_TRACE_KFD"] = "1"
            if getattr(cfg, "pc_sampling", False):
                env["SOFA_PC_SAMPLING"] = "1"
        else:
            p.print_warning("libsofatracer.so not built; GPU tracing disabled")
        if cfg.rccl_shim:
            shim = native_lib("libsofarccl.so")
            if os.path.exists(shim):
                prev_ld = env.get("LD_PRELOAD", "")
                env["LD_PRELOAD"] = shim + ((":" + prev_ld) if prev_ld else "")
            else:
                p.print_warning("libsofarccl.so not built; RCCL shim disabled")
    return env


def sofa_attach(pid: int, cfg: SofaConfig, duration: float = 0.0) -> int:
    """Observe an already-running process: CPU sampling + system monitors
    (the GPU collector cannot attach post-init — rocprofiler tools load with
    the runtime — so GPU activity comes from telemetry only in this mode).

    Runs until the target exits, `duration` elapses, or Ctrl-C.
    """
    logdir = ensure_logdir(cfg)
    ensure_native_built(cfg.verbose)
    sofa_clean(cfg)
    if not os.path.exists(f"/proc/{pid}"):
        p.print_error(f"pid {pid} does not exist")
        return 2
    p.print_progress(f"attaching to pid {pid} (logdir {logdir})")

    with open(os.path.join(logdir, "sofa_time.txt"), "w") as f:
        f.write("%.9f\n" % time.time())
    _write_timebase(logdir)
    try:
        shutil.copyfile("/proc/kallsyms", os.path.join(logdir, "kallsyms"))
    except OSError:
        pass
    if cfg.enable_gpu:
        dump_xgmi_topology(logdir)

    mon = SysMonitor(logdir, rate_hz=cfg.sys_mon_rate, enable_gpu=cfg.enable_gpu)
    mon.start()
    sampler = None
    sampler_bin = native_bin("sofa-cpusampler")
    if os.path.exists(sampler_bin):
        args = [sampler_bin, "-o", os.path.join(logdir, "cpusamples.scs"),
                "-F", str(cfg.cpu_sample_rate), "-p", str(pid)]
        if cfg.enable_callchain:
            args.append("-g")
        sampler = subprocess.Popen(args)

    t_begin = time.time()
    ret = 0
    try:
        while os.path.exists(f"/proc/{pid}"):
            time.sleep(0.2)
            if duration and time.time() - t_begin >= duration:
                break
    except KeyboardInterrupt:
        pass
    t_end = time.time()

    if sampler is not None:
        try:
            sampler.terminate()
            sampler.wait(timeout=5)
        except (OSError, subprocess.TimeoutExpired):
            sampler.kill()
    mon.stop()
    mon.join(timeout=5)
    with open(os.path.join(logdir, "misc.txt"), "w") as f:
        f.write(json.dumps({
            "elapsed_time": t_end - t_begin,
            "cores": os.cpu_count(),
            "pid": pid,
            "returncode": 0,
            "command": f"--attach {pid}",
        }))
    p.print_progress("attach recording done (%.2f s)" % (t_end - t_begin))
    return ret


def sofa_record(command: str, cfg: SofaConfig, duration: float = 0.0) -> int:
    logdir = ensure_logdir(cfg)
    ensure_native_built(cfg.verbose)
    sofa_clean(cfg)

    if getattr(cfg, "docker_image", ""):
        # container target: image CMD (or `command`) profiled inside docker
        # (reference bin/sofa_record.py:362-399)
        with open(os.path.join(logdir, "sofa_time.txt"), "w") as f:
            f.write("%.9f\n" % time.time())
        _write_timebase(logdir)
        try:
            shutil.copyfile("/proc/kallsyms", os.path.join(logdir, "kallsyms"))
        except OSError:
            pass
        from .docker_target import record_docker

        return record_docker(cfg, cfg.docker_image, command, logdir)

    p.print_progress(f"recording into {logdir}: {command}")

    # --- prologue: clock base + symbols + topology ---
    with open(os.path.join(logdir, "sofa_time.txt"), "w") as f:
        f.write("%.9f\n" % time.time())
    _write_timebase(logdir)
    try:
        shutil.copyfile("/proc/kallsyms", os.path.join(logdir, "kallsyms"))
    except OSError:
        pass
    if cfg.enable_gpu:
        dump_xgmi_topology(logdir)
        try:
            # run the timebase/MFMA-marker prologue as a SUBPROCESS carrying
            # the tracer env: the collector then records the
            # mfma_marker_kernel span into the same logdir, and analyze can
            # cross-check it against the marker's s_memrealtime
            # self-measurement (on-device ground truth for the tracer clock)
            rc = subprocess.run(
                [sys.executable, "-c",
                 "from sofa_amd.record.gpu_timebase import write_gpu_timebase;"
                 f"import sys; sys.exit(0 if write_gpu_timebase({logdir!r}) else 1)"],
                env=build_target_env(cfg), timeout=120,
                cwd=os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
            ).returncode
            if rc == 0:
                p.print_info("GPU timebase + MFMA marker correlation recorded")
        except Exception as e:
            p.print_warning(f"gpu timebase prologue failed: {e}")

    # --- background monitors ---
    mon = SysMonitor(logdir, rate_hz=cfg.sys_mon_rate, enable_gpu=cfg.enable_gpu)
    mon.start()

    blk = None
    if getattr(cfg, "enable_blkio", False) or cfg.blkdev:
        from .blkio import BlkTracer

        blk = BlkTracer(logdir, device=cfg.blkdev)
        blk.start()

    pktcap_proc = None
    if cfg.enable_tcpdump:
        sniffer = native_bin("sofa-pktcap")
        if os.path.exists(sniffer):
            try:
                pktcap_proc = subprocess.Popen(
                    [sniffer, "-o", os.path.join(logdir, "pktcap.bin")],
                    stderr=subprocess.DEVNULL,
                )
            except OSError:
                pktcap_proc = None
        else:
            p.print_warning("sofa-pktcap not built; packet capture disabled")

    # --- launch target ---
    env = build_target_env(cfg)
    err_f = open(os.path.join(logdir, "sofa.err"), "w")
    argv = ["bash", "-c", command]
    if cfg.enable_strace:
        st_bin = native_bin("sofa-syscalltrace")
        if os.path.exists(st_bin):
            argv = [st_bin, "-o", os.path.join(logdir, "strace.sst"), "--"] + argv
        else:
            p.print_warning("sofa-syscalltrace not built; syscall tracing disabled")
    if cfg.enable_pystacks:
        # inject the in-process Python stack sampler via PYTHONPATH
        # (pyflame-replacement; see sofa_amd/pystacks_inject)
        inject_dir = os.path.join(os.path.dirname(native_dir()), "pystacks_inject")
        if os.path.isdir(inject_dir):
            env["PYTHONPATH"] = inject_dir + ":" + env.get("PYTHONPATH", "")
            env["SOFA_PYSTACKS_OUT"] = os.path.join(logdir, "pystacks.txt")
    t_begin = time.time()
    target = subprocess.Popen(
        argv,
        env=env,
        stdout=None,
        stderr=err_f if not cfg.verbose else None,
        start_new_session=False,
    )

    # --- CPU sampler attached to the target ---
    sampler = None
    sampler_bin = native_bin("sofa-cpusampler")
    if os.path.exists(sampler_bin):
        args = [sampler_bin, "-o", os.path.join(logdir, "cpusamples.scs"), "-F", str(cfg.cpu_sample_rate)]
        if cfg.perf_events and cfg.perf_events != "cpu-clock":
            args += ["-e", cfg.perf_events]
        if cfg.enable_callchain:
            args.append("-g")
        if cfg.profile_all_cpus or cfg.enable_strace:
            # strace mode: the real workload is a grandchild the ptrace
            # wrapper forked before the sampler attached, so per-pid inherit
            # would miss it — sample system-wide instead
            args.append("-a")
        else:
            args += ["-p", str(target.pid)]
        try:
            sampler = subprocess.Popen(args)
        except OSError as e:
            p.print_warning(f"cpusampler failed to start: {e}")
    else:
        p.print_warning("sofa-cpusampler not built; CPU sampling disabled")

    # pid inventory for tools/killsofa.sh (kill exact PIDs, never patterns)
    with open(os.path.join(logdir, "sofa_pids.txt"), "w") as f:
        f.write("%d\n" % target.pid)
        if sampler is not None:
            f.write("%d\n" % sampler.pid)
        if pktcap_proc is not None:
            f.write("%d\n" % pktcap_proc.pid)

    # --- wait (optionally time-boxed: a first-class version of the
    # reference's EDR `sofa record "sleep N"` trick) ---
    try:
        if duration > 0:
            try:
                ret = target.wait(timeout=duration)
            except subprocess.TimeoutExpired:
                p.print_progress(f"duration {duration}s reached; stopping target")
                target.terminate()
                try:
                    ret = target.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    target.kill()
                    ret = target.wait()
        else:
            ret = target.wait()
    except KeyboardInterrupt:
        target.send_signal(signal.SIGINT)
        ret = target.wait()
    t_end = time.time()

    # --- epilogue ---
    if sampler is not None:
        try:
            sampler.terminate()
            sampler.wait(timeout=5)
        except (OSError, subprocess.TimeoutExpired):
            sampler.kill()
    if pktcap_proc is not None:
        try:
            pktcap_proc.terminate()
            pktcap_proc.wait(timeout=5)
        except (OSError, subprocess.TimeoutExpired):
            pktcap_proc.kill()
    if blk is not None:
        blk.stop()
    mon.stop()
    mon.join(timeout=5)
    err_f.close()

    with open(os.path.join(logdir, "misc.txt"), "w") as f:
        f.write(
            json.dumps(
                {
                    "elapsed_time": t_end - t_begin,
                    "cores": os.cpu_count(),
                    "pid": target.pid,
                    "returncode": ret,
                    "command": command,
                }
            )
        )
    if ret != 0:
        p.print_warning(f"target exited with {ret}")
    p.print_progress("recording done (elapsed %.2f s)" % (t_end - t_begin))
    return ret
