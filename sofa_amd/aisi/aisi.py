"""AISI — automatic iteration detection from the GPU kernel stream.

Parity with reference bin/sofa_aisi.py:359-516: find
per-training-iteration boundaries with zero framework instrumentation by
mining the kernel-name token stream for a pattern repeating num_iterations
times, then profile each iteration (fw/bw/gemm/copy/collective split) and
print the compute- vs communication-bound verdict (threshold 0.15,
bin/sofa_aisi.py:503-507).

Differences by design (reference bugs NOT replicated):
* device choice: the busiest GPU, not hardcoded deviceId==1;
* pattern mining: suffix automaton (stree.py) instead of suffix tree +
  fuzzy sliding window — exact occurrence positions fall out directly.
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np
import pandas as pd

from .. import printing as p
from .stree import find_repeat_pattern, occurrences

COMM_BOUND_RATIO = 0.15


def tokenize_kernels(df_gpu: pd.DataFrame) -> Tuple[List[int], np.ndarray, np.ndarray, dict]:
    """Kernel stream of the busiest device -> (token ids, t_start, t_end)."""
    kernels = df_gpu[df_gpu["copyKind"] == 0]
    if len(kernels) == 0:
        return [], np.empty(0), np.empty(0), {}
    busiest = kernels.groupby("deviceId").size().idxmax()
    k = kernels[kernels["deviceId"] == busiest].sort_values("timestamp")
    names = k["name"].astype(str)
    # strip the [gpuN] prefix and template args so identical ops tokenize equal
    base = names.str.replace(r"^\[gpu\d+\] ", "", regex=True).str.split("<").str[0]
    cats = base.astype("category")
    tokens = cats.cat.codes.to_numpy()
    vocab = dict(enumerate(cats.cat.categories))
    ts = k["timestamp"].to_numpy()
    te = ts + k["duration"].to_numpy()
    return tokens.tolist(), ts, te, vocab


def detect_iterations(
    tokens: List[int], num_iterations: int
) -> Optional[Tuple[List[int], int]]:
    """Return (occurrence starts, pattern length) for the best pattern."""
    if not tokens:
        return None
    for tol in (0, 1, 2):
        cands = find_repeat_pattern(tokens, num_iterations, tol=tol, min_len=2)
        for start, length, cnt in cands[:50]:
            pat = tokens[start : start + length]
            if len(set(pat)) < 2:  # constant patterns are degenerate
                continue
            occ = occurrences(tokens, pat)
            if abs(len(occ) - num_iterations) <= tol and len(occ) >= 2:
                return occ, length
    return None


def iter_profile(
    df_gpu: pd.DataFrame,
    df_rccl: pd.DataFrame,
    t_begin: float,
    t_end: float,
) -> dict:
    """Per-iteration split (reference iter_profile, bin/sofa_aisi.py:21-59)."""
    win = df_gpu[(df_gpu["timestamp"] >= t_begin) & (df_gpu["timestamp"] < t_end)]
    kernels = win[win["copyKind"] == 0]
    copies = win[win["copyKind"].isin([1, 2, 8, 10])]
    names = kernels["name"].astype(str)
    lower = names.str.lower()
    fw = kernels[lower.str.contains("_fw|fwd|forward", regex=True)]["duration"].sum()
    bw = kernels[lower.str.contains("_bw|bwd|backward|grad", regex=True)]["duration"].sum()
    gemm = kernels[lower.str.contains("cijk|gemm|mfma|conv|wmma", regex=True)]["duration"].sum()
    coll = kernels[lower.str.contains("rccl|nccl|ccl", regex=True)]["duration"].sum()
    if df_rccl is not None and len(df_rccl):
        rwin = df_rccl[(df_rccl["timestamp"] >= t_begin) & (df_rccl["timestamp"] < t_end)]
        coll = max(coll, rwin["duration"].sum())
    return {
        "t_begin": t_begin,
        "t_end": t_end,
        "step_time": t_end - t_begin,
        "kernel_time": kernels["duration"].sum(),
        "fw_time": fw,
        "bw_time": bw,
        "gemm_time": gemm,
        "copy_time": copies["duration"].sum(),
        "copy_payload": int(copies["payload"].sum()),
        "coll_time": coll,
        "n_kernels": len(kernels),
    }


def tokenize_strace(df_strace: pd.DataFrame):
    """Syscall stream -> tokens (reference --aisi_via_strace path)."""
    if df_strace is None or len(df_strace) == 0:
        return [], np.empty(0), np.empty(0), {}
    d = df_strace.sort_values("timestamp")
    base = d["name"].astype(str).str.split("(").str[0]
    cats = base.astype("category")
    tokens = cats.cat.codes.to_numpy()
    ts = d["timestamp"].to_numpy()
    te = ts + d["duration"].to_numpy()
    return tokens.tolist(), ts, te, dict(enumerate(cats.cat.categories))


def sofa_aisi(logdir, cfg, df_cpu, df_gpu, df_rccl, features,
              df_strace=None) -> Optional[pd.DataFrame]:
    p.print_title("AISI — iteration detection")
    if getattr(cfg, "aisi_via_strace", False):
        tokens, ts, te, vocab = tokenize_strace(df_strace)
        if not tokens:
            p.print_warning("no syscall trace for --aisi_via_strace")
            return None
        if df_gpu is None:
            df_gpu = df_strace.iloc[0:0]
    elif df_gpu is None or len(df_gpu) == 0:
        p.print_warning("no GPU trace; AISI needs kernel records")
        return None
    else:
        tokens, ts, te, vocab = tokenize_kernels(df_gpu)
    if not tokens:
        p.print_warning("no kernels to tokenize")
        return None
    det = detect_iterations(tokens, cfg.num_iterations)
    if det is None:
        p.print_warning(
            f"no pattern repeating ~{cfg.num_iterations}x found over {len(tokens)} kernels"
        )
        return None
    occ, plen = det
    rows = []
    for i, s in enumerate(occ):
        t_begin = ts[s]
        # an iteration spans from this occurrence to the next one — the mined
        # pattern may cover only a stable subsequence of the step (autotuned
        # kernels vary run to run), but occurrence STARTS delimit full steps
        if i + 1 < len(occ):
            t_end = ts[occ[i + 1]]
        else:
            t_end = te[min(s + plen - 1, len(te) - 1)]
        rows.append(iter_profile(df_gpu, df_rccl, t_begin, t_end))
    idf = pd.DataFrame(rows)

    print(
        "detected %d iterations (pattern length %d kernels); mean step %.4f s"
        % (len(idf), plen, idf["step_time"].mean())
    )
    # expose the mined kernel-name sequence itself (round-1 roadmap: the
    # pattern was detected but never shown); full sequence -> artifact,
    # compressed head -> console
    try:
        pat_tokens = tokens[occ[0] : occ[0] + plen]
        pat_names = [vocab.get(t, str(t)) for t in pat_tokens]
        with open(os.path.join(logdir, "iteration_pattern.txt"), "w") as f:
            for i, nm in enumerate(pat_names):
                f.write("%4d %s\n" % (i, nm))
        shown = [nm if len(nm) < 60 else nm[:57] + "..." for nm in pat_names[:12]]
        print("iteration pattern (first %d of %d kernels; full list in "
              "iteration_pattern.txt):" % (len(shown), plen))
        for nm in shown:
            print("    " + nm)
    except (OSError, IndexError):
        pass
    print(
        idf[["t_begin", "step_time", "fw_time", "bw_time", "gemm_time", "copy_time", "coll_time"]]
        .describe()
        .loc[["mean", "50%", "min", "max"]]
        .to_string()
    )

    features.append(("iter_count", float(len(idf))))
    features.append(("iter_step_time", float(idf["step_time"].mean())))
    features.append(("iter_fw_time", float(idf["fw_time"].mean())))
    features.append(("iter_bw_time", float(idf["bw_time"].mean())))
    features.append(("iter_copy_time", float(idf["copy_time"].mean())))
    features.append(("iter_coll_time", float(idf["coll_time"].mean())))

    comm_time = idf["copy_time"].mean() + idf["coll_time"].mean()
    ratio = comm_time / max(idf["step_time"].mean(), 1e-12)
    if ratio > COMM_BOUND_RATIO:
        p.print_hint(
            "workload looks COMMUNICATION-bound (comm/step = %.2f > %.2f)"
            % (ratio, COMM_BOUND_RATIO)
        )
    else:
        p.print_hint(
            "workload looks COMPUTE-bound (comm/step = %.2f <= %.2f)"
            % (ratio, COMM_BOUND_RATIO)
        )

    # artifacts: iteration_timeline.txt + marker series appended to report.js
    with open(os.path.join(logdir, "iteration_timeline.txt"), "w") as f:
        for i, r in idf.iterrows():
            f.write("iteration %d: %.6f .. %.6f (%.6f s)\n" % (i, r["t_begin"], r["t_end"], r["step_time"]))
    report_js = os.path.join(logdir, "report.js")
    if os.path.isfile(report_js):
        import json

        pts = [
            {"x": round(float(r["t_begin"]), 6), "y": 1.0, "name": "iter_%d" % i}
            for i, r in idf.iterrows()
        ]
        with open(report_js, "a") as f:
            f.write(
                "iteration_markers = %s;\n"
                % json.dumps({"name": "iteration begins", "color": "black", "data": pts})
            )
            f.write("sofa_traces.push(iteration_markers);\n")
    return idf
