"""Container-target profiling (docker/podman).

Reference behavior (cyliustack/sofa bin/sofa_record.py:362-399 +
bin/sofa_preprocess.py:396-414): introspect the image CMD, relaunch the
container with the logdir as a volume, profile via cgroup-scoped perf, and
resolve in-container symbols via a bindfs symfs mount.

MI355X-native redesign:
  * the container gets the logdir volume (/sofa_log), the native tracer libs
    volume (/sofa_native, read-only) and the ROCm devices
    (--device /dev/kfd /dev/dri) so the GPU collector runs INSIDE the
    container — no in-container install needed;
  * CPU sampling is cgroup-scoped from the HOST: sofa-cpusampler -G
    <cgroup dir> (perf_event_open PERF_FLAG_PID_CGROUP), no perf binary;
  * symbols: instead of bindfs, the container's merged overlayfs root
    (docker inspect .GraphDriver.Data.MergedDir) is recorded to
    container_root.txt; the Symbolizer prefixes it when a DSO path from an
    in-container mmap record does not exist on the host.

SOFA_DOCKER_BIN overrides the docker binary (tests use a stub)."""

from __future__ import annotations

import json
import os
import subprocess
import time
from typing import List, Optional, Tuple

from .. import printing as p

CGROUP_BASES = (
    "/sys/fs/cgroup/system.slice/docker-{cid}.scope",          # cgroup v2
    "/sys/fs/cgroup/docker/{cid}",                             # cgroupfs v1/v2
    "/sys/fs/cgroup/perf_event/docker/{cid}",                  # v1 perf ctrl
    "/sys/fs/cgroup/perf_event/system.slice/docker-{cid}.scope",
)


def docker_bin() -> str:
    return os.environ.get("SOFA_DOCKER_BIN", "docker")


def image_cmd(image: str) -> List[str]:
    """The image's default CMD (reference `docker create` introspection)."""
    out = subprocess.check_output(
        [docker_bin(), "inspect", "-f", "{{json .Config.Cmd}}", image],
        text=True,
    ).strip()
    cmd = json.loads(out) if out and out != "null" else []
    return cmd or ["sh"]


def find_cgroup(cid: str) -> Optional[str]:
    for pat in CGROUP_BASES:
        path = pat.format(cid=cid)
        if os.path.isdir(path):
            return path
    return None


def container_merged_root(cid: str) -> str:
    try:
        out = subprocess.check_output(
            [docker_bin(), "inspect", "-f", "{{.GraphDriver.Data.MergedDir}}", cid],
            text=True,
        ).strip()
        return "" if out == "<no value>" else out
    except (subprocess.CalledProcessError, OSError):
        return ""


def launch_container(
    image: str,
    command: List[str],
    logdir: str,
    gpu_env: dict,
    extra_args: Optional[List[str]] = None,
) -> Tuple[subprocess.Popen, str]:
    """Start the profiled container; returns (proc, container_id)."""
    logdir = os.path.abspath(logdir)
    native_dir = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "native"
    )
    cidfile = os.path.join(logdir, "cidfile.txt")
    if os.path.exists(cidfile):
        os.unlink(cidfile)
    args = [
        docker_bin(), "run", "--rm",
        "--cidfile", cidfile,
        "-v", f"{logdir}:/sofa_log",
        "-v", f"{native_dir}:/sofa_native:ro",
        "-e", "SOFA_LOGDIR=/sofa_log",
    ]
    # GPU collector env, rewritten to the in-container mount points
    for key, val in (gpu_env or {}).items():
        args += ["-e", f"{key}={val}"]
    if os.path.exists("/dev/kfd"):
        args += ["--device", "/dev/kfd", "--device", "/dev/dri",
                 "--group-add", "video", "--security-opt", "seccomp=unconfined"]
    args += extra_args or []
    args.append(image)
    args += command
    p.print_progress("docker: " + " ".join(args))
    proc = subprocess.Popen(args)
    # wait for the container id
    cid = ""
    for _ in range(300):
        if os.path.exists(cidfile):
            with open(cidfile) as f:
                cid = f.read().strip()
            if cid:
                break
        if proc.poll() is not None:
            break
        time.sleep(0.1)
    return proc, cid


def container_gpu_env(mode: str = "sdk") -> dict:
    """Tracer env pointing at the in-container /sofa_native mount."""
    env = {}
    if mode == "lite":
        env["HSA_TOOLS_LIB"] = "/sofa_native/lib/libsofahsalite.so"
        env["ROCP_TOOL_LIBRARIES"] = "/sofa_native/lib/libsofatracer.so"
        env["SOFA_TRACE_DISPATCH"] = "0"
        env["SOFA_TRACE_COPY"] = "1"
    else:
        env["ROCP_TOOL_LIBRARIES"] = "/sofa_native/lib/libsofatracer.so"
    return env


def record_docker(cfg, image: str, command_str: str, logdir: str) -> int:
    """Profile an image's CMD (or an explicit command) inside a container."""
    from .pollers import SysMonitor
    from .recorder import native_bin

    command = command_str.split() if command_str else image_cmd(image)
    mon = SysMonitor(logdir, rate_hz=cfg.sys_mon_rate, enable_gpu=cfg.enable_gpu)
    mon.start()
    proc, cid = launch_container(
        image, command, logdir,
        container_gpu_env(getattr(cfg, "gpu_tracer", "sdk")) if cfg.enable_gpu else {},
    )
    sampler = None
    if cid:
        p.print_info(f"container {cid[:12]} started")
        root = container_merged_root(cid)
        if root:
            with open(os.path.join(logdir, "container_root.txt"), "w") as f:
                f.write(root + "\n")
        cg = find_cgroup(cid)
        if cg:
            sampler_bin = native_bin("sofa-cpusampler")
            if os.path.exists(sampler_bin):
                sampler = subprocess.Popen(
                    [sampler_bin, "-o", os.path.join(logdir, "cpusamples.scs"),
                     "-F", str(cfg.cpu_sample_rate), "-G", cg]
                )
                p.print_info(f"cgroup-scoped CPU sampling: {cg}")
        else:
            p.print_warning("container cgroup not found; CPU sampling skipped")
    else:
        p.print_warning("no container id (docker failed to start?)")
    rc = proc.wait()
    if sampler is not None:
        sampler.terminate()
        try:
            sampler.wait(timeout=5)
        except subprocess.TimeoutExpired:
            sampler.kill()
    mon.stop()
    mon.join(timeout=5)
    return rc
