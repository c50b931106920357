#!/usr/bin/env python3
"""Multi-distro e2e matrix harness (reference test/test.py:28-78 parity).

For each test_matrix/Dockerfile.<distro>: build an image with the repo
copied in, run the real CLI inside (`sofa stat "sleep 2"` then
`sofa report --verbose`), and PASS iff stdout contains the `Complete!!`
sentinel (printed by sofa_amd.analyze at the end of a good run — the same
check the reference used).  The containers have no GPU: this IS the tested
degradation path (every GPU/net/disk stream must be optional).

Requires docker (or SOFA_DOCKER_BIN override; the unit tests drive this
harness with a stub).  Results append to test_matrix/results.log.

Usage: python tools/test_matrix.py [distro ...]
"""

from __future__ import annotations

import datetime
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MATRIX_DIR = os.path.join(REPO, "test_matrix")


def docker_bin() -> str:
    return os.environ.get("SOFA_DOCKER_BIN", "docker")


def distros():
    return sorted(
        f.split(".", 1)[1]
        for f in os.listdir(MATRIX_DIR)
        if f.startswith("Dockerfile.")
    )


def run_one(distro: str, timeout: int = 1800) -> bool:
    tag = f"sofa-amd-test-{distro}"
    df = os.path.join(MATRIX_DIR, f"Dockerfile.{distro}")
    build = subprocess.run(
        [docker_bin(), "build", "-f", df, "-t", tag, REPO],
        capture_output=True, text=True, timeout=timeout,
    )
    if build.returncode != 0:
        print(f"[{distro}] BUILD FAILED:\n{build.stderr[-1500:]}")
        return False
    run = subprocess.run(
        [docker_bin(), "run", "--rm", tag, "bash", "-c",
         'sofa stat "sleep 2" --logdir /tmp/log && '
         "sofa report --logdir /tmp/log --verbose"],
        capture_output=True, text=True, timeout=timeout,
    )
    ok = run.returncode == 0 and "Complete!!" in run.stdout
    if not ok:
        print(f"[{distro}] RUN FAILED (rc={run.returncode}):\n"
              f"{run.stdout[-1500:]}\n{run.stderr[-800:]}")
    return ok


def main() -> int:
    targets = sys.argv[1:] or distros()
    results = []
    for distro in targets:
        ok = run_one(distro)
        results.append((distro, ok))
        print(f"[{distro}] {'PASSED' if ok else 'FAILED'}")
    stamp = datetime.datetime.now().isoformat(timespec="seconds")
    with open(os.path.join(MATRIX_DIR, "results.log"), "a") as f:
        for distro, ok in results:
            f.write(f"{stamp} {distro} {'PASSED' if ok else 'FAILED'}\n")
    return 0 if all(ok for _, ok in results) else 1


if __name__ == "__main__":
    sys.exit(main())
