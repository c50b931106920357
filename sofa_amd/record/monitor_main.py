"""Standalone SysMonitor process: `python -m sofa_amd.record.monitor_main
--logdir D --rate 10 [--no-gpu] [--parent PID]`.

Used by bench.py so telemetry polling runs outside the workload process
(no GIL sharing), matching the real `sofa record` architecture where the
monitor lives in the recorder process.
"""

from __future__ import annotations

import argparse
import os
import signal
import sys
import time

from .pollers import SysMonitor


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--logdir", required=True)
    ap.add_argument("--rate", type=float, default=10.0)
    ap.add_argument("--no-gpu", action="store_true")
    ap.add_argument("--parent", type=int, default=0)
    args = ap.parse_args()

    mon = SysMonitor(args.logdir, rate_hz=args.rate, enable_gpu=not args.no_gpu)
    mon.start()

    stop = {"flag": False}

    def on_sig(*_):
        stop["flag"] = True

    signal.signal(signal.SIGTERM, on_sig)
    signal.signal(signal.SIGINT, on_sig)
    while not stop["flag"]:
        time.sleep(0.2)
        if args.parent and not os.path.exists(f"/proc/{args.parent}"):
            break
    mon.stop()
    mon.join(timeout=5)
    return 0


if __name__ == "__main__":
    sys.exit(main())
