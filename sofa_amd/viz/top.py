"""sofa top — live combined CPU/GPU/NIC/disk monitor (nvidia-smi dmon-style,
but whole-system).  Uses the same sources as the recorder's SysMonitor:
/proc snapshots + librocm_smi64 — no subprocesses.

    sofa top [--interval 1.0] [--once]
"""

from __future__ import annotations

import os
import sys
import time

from ..record import pollers
from ..record.rocsmi import RocmSmi


def _bar(pct: float, width: int = 24) -> str:
    pct = max(0.0, min(100.0, pct))
    fill = int(pct / 100 * width)
    return "[" + "#" * fill + "-" * (width - fill) + "] %5.1f%%" % pct


def snapshot(prev: dict) -> tuple:
    cur = {
        "t": time.time(),
        "stat": pollers.read_proc_stat_cpus(),
        "net": pollers.read_netdev(),
        "disk": pollers.read_diskstats(),
    }
    lines = []
    if prev:
        dt = max(cur["t"] - prev["t"], 1e-3)
        # cpu
        prev_by = {r[0]: r for r in prev["stat"]}
        busy_sum, n = 0.0, 0
        for row in cur["stat"]:
            p0 = prev_by.get(row[0])
            if not p0:
                continue
            d = [a - b for a, b in zip(row[1:], p0[1:])]
            tot = sum(d)
            if tot > 0:
                busy_sum += 100.0 * (tot - d[3]) / tot  # idle is index 3
                n += 1
        if n:
            lines.append("CPU  (%3d cores) %s" % (n, _bar(busy_sum / n)))
        # net
        pn = {r[0]: r for r in prev["net"]}
        rx = tx = 0
        for r in cur["net"]:
            if r[0] == "lo" or r[0] not in pn:
                continue
            rx += r[1] - pn[r[0]][1]
            tx += r[3] - pn[r[0]][3]
        lines.append("NIC  rx %8.2f MB/s   tx %8.2f MB/s" % (rx / dt / 1e6, tx / dt / 1e6))
        # disk
        pdk = {r[0]: r for r in prev["disk"]}
        rd = wr = 0
        for r in cur["disk"]:
            if r[0] not in pdk:
                continue
            rd += (r[2] - pdk[r[0]][2]) * 512
            wr += (r[5] - pdk[r[0]][5]) * 512
        lines.append("DISK rd %8.2f MB/s   wr %8.2f MB/s" % (rd / dt / 1e6, wr / dt / 1e6))
    return cur, lines


def sofa_top(interval: float = 1.0, once: bool = False) -> int:
    smi = None
    try:
        smi = RocmSmi()
    except Exception:
        pass
    have_gpu = smi is not None and smi.available

    prev: dict = {}
    try:
        while True:
            prev, lines = snapshot(prev)
            if lines or once:
                if not once:
                    sys.stdout.write("\033[2J\033[H")
                print("sofa top — %s" % time.strftime("%H:%M:%S"))
                for ln in lines:
                    print(ln)
                if have_gpu:
                    for dev in range(smi.n_devices):
                        busy = smi.busy_percent(dev) or 0
                        vram = (smi.memory_usage(dev) or 0) / 1e9
                        power = smi.power_watts(dev) or 0.0
                        gm = smi.gpu_metrics(dev)
                        mm = gm["mm_activity"] if gm else -1
                        mm_s = (" mm %3d%%" % mm) if mm >= 0 else ""
                        print(
                            "GPU%-2d %s  vram %6.1f GB  %5.0f W%s"
                            % (dev, _bar(float(busy)), vram, power, mm_s)
                        )
                elif once:
                    print("(no GPU visible)")
                sys.stdout.flush()
            if once and lines:
                break
            time.sleep(interval)
    except KeyboardInterrupt:
        pass
    finally:
        if smi is not None:
            smi.shutdown()
    return 0
