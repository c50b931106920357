"""System-telemetry pollers.

Replaces the reference's per-stream daemon threads + external tools
(cyliustack/sofa bin/sofa_record.py:249-312: vmstat subprocess, cpuinfo/
mpstat/diskstat/netstat /proc readers, nvidia-smi dmon/query) with one
polling thread sampling every source per tick at ``cfg.sys_mon_rate`` Hz —
consistent timestamps across streams, no subprocesses.

Raw-file formats are simple whitespace TSVs with a leading epoch timestamp;
parsed vectorized by sofa_amd.preprocess.sysmon.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Dict, List, Optional, TextIO

from .rocsmi import RocmSmi


def read_proc_stat_cpus() -> List[List[int]]:
    """Per-cpu jiffy counters [cpu_idx, usr, nice, sys, idle, iow, irq, sirq, steal]."""
    rows = []
    with open("/proc/stat") as f:
        for line in f:
            if line.startswith("cpu") and line[3] != " ":
                parts = line.split()
                idx = int(parts[0][3:])
                vals = [int(x) for x in parts[1:9]]
                rows.append([idx] + vals)
    return rows


def read_cpuinfo_mhz() -> List[float]:
    mhz = []
    try:
        with open("/proc/cpuinfo") as f:
            for line in f:
                if line.startswith("cpu MHz"):
                    mhz.append(float(line.split(":")[1]))
    except (OSError, ValueError):
        pass
    return mhz


def read_diskstats() -> List[List]:
    rows = []
    try:
        with open("/proc/diskstats") as f:
            for line in f:
                p = line.split()
                if len(p) < 14:
                    continue
                name = p[2]
                # skip partitions-of and loop/ram devices
                if name.startswith(("loop", "ram", "zram")):
                    continue
                # reads, sectors_read, ms_read, writes, sectors_written, ms_write, inflight
                rows.append(
                    [
                        name,
                        int(p[3]),
                        int(p[5]),
                        int(p[6]),
                        int(p[7]),
                        int(p[9]),
                        int(p[10]),
                        int(p[11]),
                    ]
                )
    except OSError:
        pass
    return rows


def read_netdev() -> List[List]:
    rows = []
    try:
        with open("/proc/net/dev") as f:
            for line in f.readlines()[2:]:
                iface, rest = line.split(":", 1)
                iface = iface.strip()
                p = rest.split()
                # rx_bytes rx_pkts ... tx_bytes tx_pkts
                rows.append([iface, int(p[0]), int(p[1]), int(p[8]), int(p[9])])
    except OSError:
        pass
    return rows


def read_vmstat() -> Dict[str, int]:
    out: Dict[str, int] = {}
    try:
        with open("/proc/vmstat") as f:
            for line in f:
                k, v = line.split()
                if k in ("pgpgin", "pgpgout", "pswpin", "pswpout"):
                    out[k] = int(v)
        with open("/proc/stat") as f:
            for line in f:
                p = line.split()
                if p[0] in ("ctxt", "intr", "procs_running", "procs_blocked"):
                    out[p[0]] = int(p[1])
    except OSError:
        pass
    return out


class SysMonitor(threading.Thread):
    """One thread, all telemetry streams, one timestamp per tick."""

    def __init__(self, logdir: str, rate_hz: float = 10.0, enable_gpu: bool = True):
        super().__init__(daemon=True)
        self.logdir = logdir
        self.period = 1.0 / max(rate_hz, 0.1)
        self.stop_event = threading.Event()
        self.enable_gpu = enable_gpu
        self.smi: Optional[RocmSmi] = None
        self._files: Dict[str, TextIO] = {}

    def _open(self, name: str) -> TextIO:
        f = open(os.path.join(self.logdir, name), "w")
        self._files[name] = f
        return f

    def run(self) -> None:
        f_cpuinfo = self._open("cpuinfo.txt")
        f_mpstat = self._open("mpstat.txt")
        f_disk = self._open("diskstat.txt")
        f_net = self._open("netstat.txt")
        f_vm = self._open("vmstat.txt")
        f_gpu = self._open("gpusmi.txt")
        f_xgmi = self._open("xgmi_counters.txt")
        if self.enable_gpu:
            try:
                self.smi = RocmSmi()
            except Exception:
                self.smi = None
        use_gpu = self.smi is not None and self.smi.available

        def tick():
            ts = time.time()
            mhz = read_cpuinfo_mhz()
            if mhz:
                f_cpuinfo.write("%.6f %s\n" % (ts, " ".join("%.1f" % m for m in mhz)))
            for row in read_proc_stat_cpus():
                f_mpstat.write("%.6f %s\n" % (ts, " ".join(str(x) for x in row)))
            for row in read_diskstats():
                f_disk.write("%.6f %s %s\n" % (ts, row[0], " ".join(str(x) for x in row[1:])))
            for row in read_netdev():
                f_net.write("%.6f %s %s\n" % (ts, row[0], " ".join(str(x) for x in row[1:])))
            vm = read_vmstat()
            if vm:
                f_vm.write(
                    "%.6f %d %d %d %d %d %d %d %d\n"
                    % (
                        ts,
                        vm.get("pgpgin", 0),
                        vm.get("pgpgout", 0),
                        vm.get("pswpin", 0),
                        vm.get("pswpout", 0),
                        vm.get("ctxt", 0),
                        vm.get("intr", 0),
                        vm.get("procs_running", 0),
                        vm.get("procs_blocked", 0),
                    )
                )
            if use_gpu:
                for dev in range(self.smi.n_devices):
                    busy = self.smi.busy_percent(dev)
                    membusy = self.smi.memory_busy_percent(dev)
                    vram = self.smi.memory_usage(dev)
                    power = self.smi.power_watts(dev)
                    gm = self.smi.gpu_metrics(dev)
                    mm = gm["mm_activity"] if gm else -1
                    f_gpu.write(
                        "%.6f %d %d %d %d %.1f %d\n"
                        % (
                            ts,
                            dev,
                            busy if busy is not None else -1,
                            membusy if membusy is not None else -1,
                            vram if vram is not None else -1,
                            power if power is not None else -1.0,
                            mm,
                        )
                    )
                    # per-xGMI-link HW traffic accumulators (KB): measured
                    # ground truth the analytic ring model reconciles against
                    if gm and (any(gm["xgmi_read_kb"]) or any(gm["xgmi_write_kb"])):
                        f_xgmi.write(
                            "%.6f %d %s %s\n"
                            % (
                                ts,
                                dev,
                                " ".join(str(v) for v in gm["xgmi_read_kb"]),
                                " ".join(str(v) for v in gm["xgmi_write_kb"]),
                            )
                        )

        next_t = time.time()
        while not self.stop_event.is_set():
            tick()
            next_t += self.period
            delay = next_t - time.time()
            if delay > 0:
                self.stop_event.wait(delay)
            else:
                next_t = time.time()
        # one final sample so even sub-period runs get >=2 ticks (delta
        # parsers need two snapshots)
        tick()

        for f in self._files.values():
            try:
                f.flush()
                f.close()
            except OSError:
                pass
        if self.smi is not None:
            self.smi.shutdown()

    def stop(self) -> None:
        self.stop_event.set()


def dump_xgmi_topology(logdir: str) -> bool:
    """Write xgmi_topo.txt: the xGMI/PCIe link matrix between GPUs.

    Replaces `nvidia-smi topo -m` (reference bin/sofa_record.py:311-312);
    consumed by the ring-recommendation analyzer.
    """
    try:
        smi = RocmSmi()
    except Exception:
        return False
    if not smi.available:
        return False
    try:
        mat = smi.topology_matrix()
        import json

        with open(os.path.join(logdir, "xgmi_topo.txt"), "w") as f:
            json.dump({"n_gpus": smi.n_devices, "links": mat}, f, indent=1)
        return True
    finally:
        smi.shutdown()
