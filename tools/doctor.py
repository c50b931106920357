#!/usr/bin/env python3
"""Environment check (successor of the reference's tools/prepare.sh dep
installer — this image is offline, so verify instead of install)."""

import ctypes
import os
import shutil
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

def check(name, ok, hint=""):
    print("%-42s %s%s" % (name, "OK" if ok else "MISSING", f"  ({hint})" if (hint and not ok) else ""))
    return ok

def main():
    good = True
    good &= check("python >= 3.8", sys.version_info >= (3, 8))
    for mod in ("numpy", "pandas", "sklearn", "scipy", "grpc", "yaml"):
        try:
            __import__(mod)
            check(f"python module {mod}", True)
        except ImportError:
            good &= check(f"python module {mod}", False, "pip install " + mod)
    rocm = os.environ.get("ROCM_PATH", "/opt/rocm")
    good &= check("ROCm (hipcc)", os.path.exists(os.path.join(rocm, "bin", "hipcc")), "install ROCm >= 6.0")
    check("rocprofiler-sdk", os.path.exists(os.path.join(rocm, "lib", "librocprofiler-sdk.so")))
    try:
        ctypes.CDLL(os.path.join(rocm, "lib", "librocm_smi64.so"))
        check("librocm_smi64", True)
    except OSError:
        check("librocm_smi64", False, "GPU telemetry disabled")
    for b in ("sofa-cpusampler", "sofa-timebase", "sofa-pktcap", "sofa-syscalltrace"):
        check(f"native {b}", os.path.exists(os.path.join(REPO, "sofa_amd", "native", "bin", b)),
              "python -m sofa_amd.native.build")
    for lib in ("libsofatracer.so", "libsofahip.so"):
        check(f"native {lib}", os.path.exists(os.path.join(REPO, "sofa_amd", "native", "lib", lib)),
              "python -m sofa_amd.native.build")
    try:
        with open("/proc/sys/kernel/perf_event_paranoid") as f:
            v = int(f.read())
        check("perf_event_paranoid <= 2 or root", v <= 2 or os.geteuid() == 0, "tools/empower.py")
    except OSError:
        pass
    import torch
    check("torch", True)
    check("torch GPU available", torch.cuda.is_available(), "CPU-only: GPU streams disabled")
    return 0 if good else 1

if __name__ == "__main__":
    sys.exit(main())
