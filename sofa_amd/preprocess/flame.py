"""Fold call-stack samples into flamegraph input (flamegraph.folded).

New capability beyond the reference (its pyflame path produced Python-only
flamecharts): native user+kernel stacks from the sampler's -g mode, folded
as `root;...;leaf count` lines rendered by sofaboard/flame.html.
"""

from __future__ import annotations

import os
from collections import Counter
from typing import Optional

import numpy as np

from .scs import ScsFile
from .symbols import Symbolizer


def fold_stacks(scs: ScsFile, logdir: str = "") -> Counter:
    cs = scs.samples_cs
    if len(cs) == 0:
        return Counter()
    kallsyms = os.path.join(logdir, "kallsyms") if logdir else ""
    if kallsyms and not os.path.isfile(kallsyms):
        kallsyms = ""
    symr = Symbolizer(scs.mmaps, kallsyms)

    # resolve unique (pid, frame-addr) pairs once
    cache = {}

    def frame_name(pid: int, addr: int, kernel: bool) -> str:
        key = (pid, addr, kernel)
        nm = cache.get(key)
        if nm is None:
            sym, dso = symr.resolve(pid, addr, kernel)
            nm = sym if sym.startswith("0x") is False else f"{dso}+{sym}"
            cache[key] = nm
        return nm

    folded: Counter = Counter()
    pids = cs["pid"]
    flags = cs["flags"]
    n_frames = cs["n_frames"]
    frames = cs["frames"]
    for i in range(len(cs)):
        pid = int(pids[i])
        k = int(n_frames[i])
        if k == 0:
            continue
        kernel_sample = bool(flags[i] & 1)
        # frames are leaf-first; flamegraph wants root-first
        stack = []
        for j in range(k - 1, -1, -1):
            addr = int(frames[i][j])
            # kernel frames sit above the user/kernel boundary; a kernel-mode
            # sample's leaf frames are kernel addresses
            is_kernel = addr >= 0xFFFF000000000000
            stack.append(frame_name(pid, addr, is_kernel or (kernel_sample and j == 0)))
        comm = scs.comms.get(int(cs["tid"][i]), "") or str(pid)
        folded[comm + ";" + ";".join(stack)] += 1
    return folded


def write_folded(scs: ScsFile, logdir: str) -> Optional[str]:
    folded = fold_stacks(scs, logdir)
    if not folded:
        return None
    path = os.path.join(logdir, "flamegraph.folded")
    with open(path, "w") as f:
        for stack, count in sorted(folded.items()):
            f.write(f"{stack} {count}\n")
    return path
