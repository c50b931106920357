#!/usr/bin/env python3
"""Grant capabilities so sofa collectors work without root.

Parity with reference tools/empower.py:46-68 (sofa group + setcap on
tcpdump), adapted to the native collectors:
  * sofa-pktcap       needs cap_net_raw,cap_net_admin
  * sofa-cpusampler   needs cap_perfmon (or cap_sys_admin pre-5.8)
  * kernel.perf_event_paranoid relaxed for non-root sampling

Run as root:  python tools/empower.py [--user NAME]
"""

import argparse
import grp
import os
import pwd
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "sofa_amd", "native", "bin")


def run(cmd):
    print("+ " + " ".join(cmd))
    return subprocess.run(cmd, check=False).returncode


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--user", default=os.environ.get("SUDO_USER", ""))
    ap.add_argument("--group", default="sofa")
    args = ap.parse_args()

    if os.geteuid() != 0:
        print("must run as root")
        return 1

    try:
        grp.getgrnam(args.group)
    except KeyError:
        run(["groupadd", args.group])
    if args.user:
        try:
            pwd.getpwnam(args.user)
            run(["usermod", "-aG", args.group, args.user])
        except KeyError:
            print(f"user {args.user} not found; skipping group add")

    caps = {
        "sofa-pktcap": "cap_net_raw,cap_net_admin=eip",
        "sofa-cpusampler": "cap_perfmon,cap_sys_ptrace=eip",
        "sofa-syscalltrace": "cap_sys_ptrace=eip",
    }
    for name, cap in caps.items():
        path = os.path.join(BIN, name)
        if os.path.exists(path):
            run(["chgrp", args.group, path])
            run(["chmod", "750", path])
            run(["setcap", cap, path])

    # relax perf_event_paranoid for non-root sampling
    run(["sysctl", "-w", "kernel.perf_event_paranoid=1"])
    print("done — re-login for group membership to take effect")
    return 0


if __name__ == "__main__":
    sys.exit(main())
