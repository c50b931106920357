"""Build every native component in-tree.

Host-only pieces build with g++; device code builds with hipcc for gfx950
(MI355X) only — no multi-arch fatbins, no CUDA paths.  Outputs land in
sofa_amd/native/{bin,lib} so they travel with the repo snapshot to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
BIN = os.path.join(HERE, "bin")
LIB = os.path.join(HERE, "lib")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
GFX_ARCH = "gfx950"


def _newer(dst: str, *srcs: str) -> bool:
    if not os.path.exists(dst):
        return False
    dt = os.path.getmtime(dst)
    return all(os.path.getmtime(s) <= dt for s in srcs if os.path.exists(s))


def _run(cmd, verbose=False):
    if verbose:
        print("+ " + " ".join(cmd))
    subprocess.run(cmd, check=True)


def build_all(verbose: bool = False) -> None:
    os.makedirs(BIN, exist_ok=True)
    os.makedirs(LIB, exist_ok=True)

    host_targets = [
        (
            os.path.join(BIN, "sofa-cpusampler"),
            [os.path.join(HERE, "cpusampler", "cpusampler.cc")],
            ["g++", "-O2", "-std=c++17"],
            [],
        ),
        (
            os.path.join(BIN, "sofa-timebase"),
            [os.path.join(HERE, "timebase", "timebase.cc")],
            ["g++", "-O2", "-std=c++17"],
            [],
        ),
        (
            os.path.join(BIN, "sofa-pktcap"),
            [os.path.join(HERE, "pktcap", "pktcap.cc")],
            ["g++", "-O2", "-std=c++17"],
            [],
        ),
        (
            os.path.join(BIN, "sofa-syscalltrace"),
            [os.path.join(HERE, "syscalltrace", "syscalltrace.cc")],
            ["g++", "-O2", "-std=c++17"],
            [],
        ),
        (
            os.path.join(LIB, "libsofarccl.so"),
            [
                os.path.join(HERE, "rccl_shim", "rccl_shim.cc"),
                os.path.join(HERE, "collector", "sgt_format.h"),
            ],
            ["g++", "-O2", "-std=c++17", "-fPIC", "-shared"],
            ["-ldl", "-lpthread"],
        ),
        (
            os.path.join(LIB, "libsofatracer.so"),
            [
                os.path.join(HERE, "collector", "sofatracer.cc"),
                os.path.join(HERE, "collector", "sgt_format.h"),
            ],
            [
                "g++",
                "-O2",
                "-std=c++17",
                "-fPIC",
                "-shared",
                "-D__HIP_PLATFORM_AMD__",
                f"-I{ROCM}/include",
            ],
            [f"-L{ROCM}/lib", "-lrocprofiler-sdk", f"-Wl,-rpath,{ROCM}/lib"],
        ),
        (
            os.path.join(LIB, "libsofahsalite.so"),
            [
                os.path.join(HERE, "hsalite", "hsalite.cc"),
                os.path.join(HERE, "collector", "sgt_format.h"),
            ],
            [
                "g++",
                "-O2",
                "-g",  # line info for crash triage; free at runtime
                "-std=c++17",
                "-fPIC",
                "-shared",
                # hsa_api_trace.h uses the in-tree "inc/" include layout
                # unless AMD_INTERNAL_BUILD; the installed headers are flat
                "-DAMD_INTERNAL_BUILD",
                f"-I{ROCM}/include",
                f"-I{ROCM}/include/hsa",
            ],
            ["-lpthread"],
        ),
    ]
    for dst, srcs, cc, link in host_targets:
        if not os.path.exists(srcs[0]):
            continue
        if _newer(dst, *srcs):
            continue
        cc_srcs = [s for s in srcs if s.endswith((".cc", ".cpp"))]
        _run(cc + ["-o", dst] + cc_srcs + link, verbose)

    # --- HIP device code (gfx950) ---
    hipcc = os.path.join(ROCM, "bin", "hipcc")
    ring_h = os.path.join(HERE, "hip", "trace_ring.h")
    hip_targets = [
        (
            os.path.join(LIB, "libsofahip.so"),
            [
                os.path.join(HERE, "hip", "timebase_kernel.hip"),
                os.path.join(HERE, "hip", "trace_ring.hip"),
                os.path.join(HERE, "hip", "ring_writer.hip"),
                ring_h,
            ],
        ),
        (
            os.path.join(BIN, "sofa-bandwidth"),
            [os.path.join(HERE, "hip", "bandwidth.hip"), ring_h],
        ),
    ]
    for dst, srcs in hip_targets:
        srcs = [s for s in srcs if os.path.exists(s)]
        if not srcs:
            continue
        if _newer(dst, *srcs):
            continue
        cmd = [
            hipcc,
            f"--offload-arch={GFX_ARCH}",
            "-O3",
            "-std=c++17",
        ]
        if dst.endswith(".so"):
            cmd += ["-fPIC", "-shared"]
        cmd += ["-o", dst] + [s for s in srcs if s.endswith(".hip")]
        if dst.endswith("sofa-bandwidth"):
            # --ring mode links the device-trace-ring machinery
            cmd += ["-L" + LIB, "-lsofahip", "-Wl,-rpath,$ORIGIN/../lib"]
        _run(cmd, verbose)


if __name__ == "__main__":
    build_all(verbose=True)
