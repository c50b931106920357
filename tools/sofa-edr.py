#!/usr/bin/env python3
"""Event-driven recording: watch an application log, trigger a time-boxed
`sofa record` when a phase keyword appears.

Parity with reference tools/sofa-edr.py:15-46 (BWA/BQSR/HaplotypeCaller
phases), generalized: phases come from --phases keyword list.

Usage:
  python tools/sofa-edr.py --watch app.log --phases "forward,backward,eval" \
      [--duration 20] [--logdir-base edrlog]
"""

import argparse
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SOFA = os.path.join(REPO, "bin", "sofa")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--watch", required=True, help="log file to poll")
    ap.add_argument("--phases", required=True, help="comma-separated keywords")
    ap.add_argument("--duration", type=int, default=20)
    ap.add_argument("--logdir-base", default="edrlog")
    ap.add_argument("--poll", type=float, default=1.0)
    args = ap.parse_args()

    phases = [p.strip() for p in args.phases.split(",") if p.strip()]
    seen = set()
    pos = 0
    print(f"watching {args.watch} for phases {phases}")
    while len(seen) < len(phases):
        try:
            with open(args.watch) as f:
                f.seek(pos)
                new = f.read()
                pos = f.tell()
        except OSError:
            time.sleep(args.poll)
            continue
        for ph in phases:
            if ph in seen:
                continue
            if ph in new:
                seen.add(ph)
                logdir = f"{args.logdir_base}-{ph}"
                print(f"phase '{ph}' detected -> recording {args.duration}s into {logdir}")
                subprocess.run(
                    [
                        sys.executable, SOFA, "record", f"sleep {args.duration}",
                        "--logdir", logdir, "--profile_all_cpus",
                    ],
                    check=False,
                )
        time.sleep(args.poll)
    print("all phases recorded")


if __name__ == "__main__":
    main()
