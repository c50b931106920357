"""sofa CLI — verb dispatcher.

Parity with reference bin/sofa:43-376: verbs stat/record/preprocess/analyze/
viz/report/diff/clean, keyword:color filter flags, plugin loading, cluster
report loop.
"""

from __future__ import annotations

import argparse
import importlib
import os
import sys

from . import printing as p
from .config import Filter, SofaConfig


def parse_filters(spec: str):
    """'keyword:color,keyword2:color2' -> [Filter] (reference bin/sofa:258-294)."""
    out = []
    if not spec:
        return out
    for item in spec.split(","):
        item = item.strip()
        if not item:
            continue
        if ":" in item:
            kw, color = item.split(":", 1)
        else:
            kw, color = item, ""
        out.append(Filter(kw, color))
    return out


def build_parser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser(
        prog="sofa",
        description="MI355X-native whole-system profiler (record/preprocess/analyze/viz)",
    )
    ap.add_argument(
        "command_verb",
        choices=["stat", "record", "preprocess", "analyze", "report", "viz", "diff", "clean", "top"],
    )
    ap.add_argument("usr_command", nargs="?", default="", help="command to profile (record/stat)")
    ap.add_argument("--logdir", default="sofalog")
    ap.add_argument("--version", action="version", version="sofa-amd 0.1.0")
    ap.add_argument("--verbose", action="store_true")
    # record
    ap.add_argument("--cpu_sample_rate", type=int, default=99)
    ap.add_argument("--perf_events", default="cpu-clock",
                    help="sampling event: cpu-clock (default) or cycles (needs a PMU)")
    ap.add_argument("--sys_mon_rate", type=int, default=10)
    ap.add_argument("--profile_all_cpus", action="store_true")
    ap.add_argument("--attach", type=int, default=0, metavar="PID",
                    help="record an already-running process instead of launching one")
    ap.add_argument("--duration", type=float, default=0.0,
                    help="stop recording after N seconds (works for launched "
                    "commands and --attach)")
    ap.add_argument("--enable_tcpdump", action="store_true")
    ap.add_argument("--enable_strace", action="store_true")
    ap.add_argument("--enable_py_stacks", action="store_true")
    ap.add_argument("--call_stacks", action="store_true",
                    help="sample native call stacks (-g) -> flamegraph.folded + flame.html")
    ap.add_argument("--no_gpu", action="store_true", help="disable GPU tracing")
    ap.add_argument(
        "--gpu_tracer",
        choices=["sdk", "lite"],
        default="lite",
        help="lite = HSA-level dispatch tracer (3.4%% overhead, default); "
        "sdk = rocprofiler-sdk collector (full fidelity: HIP API spans, "
        "RCCL args, KFD; ~13%% overhead)",
    )
    ap.add_argument("--no_hip_api", action="store_true",
                    help="disable HIP runtime API span tracing")
    ap.add_argument(
        "--hip_api_full",
        action="store_true",
        help="trace EVERY HIP API call, not just launch/copy/sync/alloc "
        "(adds ~45%% overhead on launch-dense workloads)",
    )
    ap.add_argument("--no_rccl", action="store_true")
    ap.add_argument(
        "--rccl_shim",
        action="store_true",
        help="trace RCCL via the LD_PRELOAD interposer instead of/next to "
        "the rocprofiler-sdk path",
    )
    ap.add_argument("--gpu_buffer_mb", type=int, default=64)
    ap.add_argument("--docker", default="", dest="docker_image", metavar="IMAGE",
                    help="profile inside a container of IMAGE (command = "
                    "image CMD when omitted); cgroup-scoped CPU sampling + "
                    "GPU tracer mounted read-only")
    ap.add_argument("--enable_blkio", action="store_true",
                    help="per-request block-IO latency via tracefs "
                    "block_rq_issue/complete (blktrace parity)")
    ap.add_argument("--blkdev", default="", help="block device hint (implies --enable_blkio)")
    ap.add_argument("--gpu_sample", type=int, default=1, metavar="N",
                    help="lite collector: time every Nth dispatch "
                    "(overhead ~3.4%%/N; submit spans keep full launch "
                    "coverage; 1 = every dispatch)")
    ap.add_argument("--pc_sampling", action="store_true",
                    help="GPU program-counter sampling (instruction-level "
                    "hotspots; implies --gpu_tracer sdk; experimental SDK API)")
    ap.add_argument("--enable_kfd_trace", action="store_true",
                    help="trace KFD page-migrate/fault events (SVM memory pressure)")
    # preprocess
    ap.add_argument("--cpu_time_offset_ms", type=int, default=0)
    ap.add_argument("--plot_ratio", type=int, default=1)
    ap.add_argument("--cpu_filters", default="", help="keyword:color,...")
    ap.add_argument("--gpu_filters", default="", help="keyword:color,...")
    ap.add_argument("--net_filters", default="", help="keyword:color,...")
    ap.add_argument("--diskstat_filters", default="")
    ap.add_argument("--enable_swarms", action="store_true")
    ap.add_argument("--num_swarms", type=int, default=10)
    # analyze
    ap.add_argument("--enable_aisi", action="store_true")
    ap.add_argument("--aisi_via_strace", action="store_true")
    ap.add_argument("--num_iterations", type=int, default=20)
    ap.add_argument("--spotlight_gpu", action="store_true")
    ap.add_argument("--profile_region", default=None, help="begin,end seconds")
    ap.add_argument("--potato_server", default=os.environ.get("POTATO_SERVER_SERVICE_HOST", ""))
    ap.add_argument("--cluster_ip", default="")
    ap.add_argument("--skip_preprocess", action="store_true")
    # diff
    ap.add_argument("--base_logdir", default="sofalog-base")
    ap.add_argument("--match_logdir", default="sofalog-match")
    # viz
    ap.add_argument("--viz_port", type=int, default=8000)
    ap.add_argument("--viz_host", default="127.0.0.1",
                    help="viz bind address (default loopback; set 0.0.0.0 to expose)")
    ap.add_argument("--interval", type=float, default=1.0, help="top refresh seconds")
    ap.add_argument("--once", action="store_true", help="top: one refresh then exit")
    ap.add_argument("--with-gui", dest="with_gui", action="store_true")
    # plugins (reference bin/sofa:21,322)
    ap.add_argument("--plugins", default="", help="comma-separated module names; each must expose f(cfg)")
    # declarative config file (successor of the reference's vestigial
    # examples/conf/*.cfg libconfig surface)
    ap.add_argument("--config", default="", help="YAML file of SofaConfig fields (CLI flags win)")
    return ap


def apply_config_file(cfg: SofaConfig, path: str) -> None:
    import yaml

    with open(path) as f:
        data = yaml.safe_load(f) or {}
    for key, val in data.items():
        if key in ("cpu_filters", "gpu_filters", "net_filters", "diskstat_filters"):
            val = [Filter(**v) if isinstance(v, dict) else Filter(str(v), "") for v in val]
        if hasattr(cfg, key):
            setattr(cfg, key, val)
        else:
            p.print_warning(f"config: unknown key {key}")


def cfg_from_args(args) -> SofaConfig:
    cfg = SofaConfig(
        logdir=args.logdir,
        command=args.usr_command,
        verbose=args.verbose,
        cpu_sample_rate=args.cpu_sample_rate,
        perf_events=args.perf_events,
        sys_mon_rate=args.sys_mon_rate,
        profile_all_cpus=args.profile_all_cpus,
        enable_tcpdump=args.enable_tcpdump,
        enable_strace=args.enable_strace,
        enable_pystacks=args.enable_py_stacks,
        enable_callchain=args.call_stacks,
        enable_gpu=not args.no_gpu,
        enable_gpu_hip_api=not args.no_hip_api,
        hip_api_full=args.hip_api_full,
        gpu_tracer=("sdk" if args.pc_sampling else args.gpu_tracer),
        pc_sampling=args.pc_sampling,
        gpu_sample=args.gpu_sample,
        enable_blkio=args.enable_blkio,
        blkdev=args.blkdev,
        docker_image=args.docker_image,
        enable_rccl_trace=not args.no_rccl,
        rccl_shim=args.rccl_shim,
        gpu_ring_buffer_mb=args.gpu_buffer_mb,
        enable_kfd_trace=args.enable_kfd_trace,
        cpu_time_offset_ms=args.cpu_time_offset_ms,
        plot_ratio=args.plot_ratio,
        cpu_filters=parse_filters(args.cpu_filters),
        net_filters=parse_filters(args.net_filters),
        diskstat_filters=parse_filters(args.diskstat_filters),
        enable_swarms=args.enable_swarms,
        num_swarms=args.num_swarms,
        enable_aisi=args.enable_aisi,
        aisi_via_strace=args.aisi_via_strace,
        num_iterations=args.num_iterations,
        spotlight_gpu=args.spotlight_gpu,
        profile_region=args.profile_region,
        potato_server=args.potato_server,
        cluster_ip=args.cluster_ip,
        skip_preprocess=args.skip_preprocess,
        base_logdir=args.base_logdir,
        match_logdir=args.match_logdir,
        viz_port=args.viz_port,
        viz_host=args.viz_host,
    )
    if args.gpu_filters:
        cfg.gpu_filters = parse_filters(args.gpu_filters)
    p.set_verbose(args.verbose)
    return cfg


def run_plugins(args, cfg) -> None:
    for name in (args.plugins or "").split(","):
        name = name.strip()
        if not name:
            continue
        try:
            mod = importlib.import_module(name)
            fn = getattr(mod, name, None) or getattr(mod, "f", None)
            if fn:
                fn(cfg)
        except Exception as e:
            p.print_warning(f"plugin {name} failed: {e}")


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    cfg = cfg_from_args(args)
    if args.config:
        try:
            apply_config_file(cfg, args.config)
        except Exception as e:
            p.print_error(f"bad --config {args.config}: {e}")
            return 2
    run_plugins(args, cfg)
    verb = args.command_verb

    if verb == "top":
        from .viz.top import sofa_top

        return sofa_top(interval=args.interval, once=args.once)

    from .record import sofa_clean, sofa_record

    if verb == "clean":
        sofa_clean(cfg)
        return 0

    if verb in ("record", "stat"):
        if args.attach:
            from .record.recorder import sofa_attach

            sofa_attach(args.attach, cfg, duration=args.duration)
        elif not args.usr_command:
            p.print_error("record/stat needs a command to profile (or --attach PID)")
            return 2
        else:
            sofa_record(args.usr_command, cfg, duration=args.duration)
        if verb == "record":
            return 0

    from .preprocess import sofa_preprocess
    from .analyze import cluster_analyze, sofa_analyze

    if verb == "diff":
        from .ml.diff import sofa_swarm_diff

        if not cfg.skip_preprocess:
            for d in (cfg.base_logdir, cfg.match_logdir):
                c2 = cfg_from_args(args)
                c2.logdir = d
                c2.enable_swarms = True
                sofa_preprocess(c2)
        sofa_swarm_diff(cfg)
        return 0

    if verb in ("stat", "report", "preprocess", "analyze"):
        if cfg.cluster_ip and verb in ("report", "analyze"):
            # cluster mode: per-node logdirs <logdir>-<ip>/ (reference
            # bin/sofa:358-367; arity bug there NOT replicated)
            base = cfg.logdir.rstrip("/")
            results = {}
            for ip in cfg.cluster_ip.split(","):
                ip = ip.strip()
                c2 = cfg_from_args(args)
                c2.logdir = f"{base}-{ip}/"
                if not cfg.skip_preprocess:
                    sofa_preprocess(c2)
                results[ip] = c2
            cluster_analyze(cfg, results)
            return 0
        pre = {}
        if verb in ("stat", "report", "preprocess") or (
            verb == "analyze" and not cfg.skip_preprocess
        ):
            pre = sofa_preprocess(cfg)
        if verb in ("stat", "report", "analyze"):
            sofa_analyze(cfg, pre)
        if verb == "report" and args.with_gui:
            from .viz import sofa_viz

            sofa_viz(cfg)
        return 0

    if verb == "viz":
        from .viz import sofa_viz

        sofa_viz(cfg)
        return 0

    return 0


if __name__ == "__main__":
    sys.exit(main())
