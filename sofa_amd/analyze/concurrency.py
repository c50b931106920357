"""Concurrency breakdown: dominant-resource attribution per time window.

Parity: reference bin/sofa_analyze.py:75-243 — windows of 1/sys_mon_rate s;
each window is attributed to the dominant of {usr, sys, gpu, iow};
Pearson correlation matrix between resource vectors; performance.csv.
Vectorized here (the reference's window loop is a known hot spot,
SURVEY.md §3.2).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np
import pandas as pd


def concurrency_breakdown(
    logdir: str,
    df_mpstat: pd.DataFrame,
    df_gpusmi: pd.DataFrame,
    df_netstat: pd.DataFrame,
    features: List[Tuple[str, float]],
    window_s: float = 0.1,
) -> Optional[pd.DataFrame]:
    if df_mpstat is None or len(df_mpstat) == 0:
        return None
    t0 = df_mpstat["timestamp"].min()
    t1 = df_mpstat["timestamp"].max()
    if t1 <= t0:
        t1 = t0 + window_s  # sub-period run: one window still gets attributed
    edges = np.arange(t0, t1 + window_s, window_s)
    n_win = len(edges) - 1
    if n_win <= 0:
        return None

    def bin_mean(df, value_col, parse=None):
        out = np.zeros(n_win)
        if df is None or len(df) == 0:
            return out
        idx = np.clip(np.searchsorted(edges, df["timestamp"].to_numpy(), "right") - 1, 0, n_win - 1)
        vals = df[value_col].to_numpy(dtype=np.float64) if parse is None else parse(df)
        sums = np.bincount(idx, weights=vals, minlength=n_win)
        counts = np.bincount(idx, minlength=n_win)
        with np.errstate(divide="ignore", invalid="ignore"):
            out = np.where(counts > 0, sums / np.maximum(counts, 1), 0.0)
        return out

    # usr/sys/iow from the mpstat trace names is lossy; recompute from CSV
    mp_csv = os.path.join(logdir, "mpstat.csv")
    usr = sysv = iow = idl = None
    if os.path.isfile(mp_csv):
        try:
            mp = pd.read_csv(mp_csv)
            usr = bin_mean(mp, "usr_r")
            sysv = bin_mean(mp, "sys_r")
            iow = bin_mean(mp, "iow_r")
            idl = bin_mean(mp, "idle_r")
        except (OSError, KeyError, ValueError):
            pass
    if usr is None:
        busy = bin_mean(df_mpstat, "duration")
        usr, sysv, iow, idl = busy, np.zeros(n_win), np.zeros(n_win), 100.0 - busy

    gpu = bin_mean(df_gpusmi, "duration") if df_gpusmi is not None else np.zeros(n_win)
    net = bin_mean(df_netstat, "bandwidth") if df_netstat is not None else np.zeros(n_win)

    stack = np.vstack([usr, sysv, gpu, iow])
    dom = np.argmax(stack, axis=0)
    active = stack.max(axis=0) > 1.0  # >1% of something happening
    names = ["usr", "sys", "gpu", "iow"]
    total_active = active.sum()
    print("\nConcurrency breakdown (dominant resource per %.2fs window):" % window_s)
    for i, nm in enumerate(names):
        frac = ((dom == i) & active).sum() / max(total_active, 1)
        print("  %-4s dominant: %5.1f%%" % (nm, 100.0 * frac))
        features.append((f"dominant_{nm}_ratio", float(frac)))

    perf = pd.DataFrame(
        {
            "t": edges[:-1],
            "usr": usr,
            "sys": sysv,
            "gpu": gpu,
            "iow": iow,
            "idl": idl,
            "net_Bps": net,
        }
    )
    perf.to_csv(os.path.join(logdir, "performance.csv"), index=False)

    # correlation matrix (reference :236-241)
    cols = ["usr", "sys", "gpu", "iow", "idl", "net_Bps"]
    with np.errstate(invalid="ignore"):
        corr = perf[cols].corr()
    print("  correlations (usr vs gpu: %.2f, gpu vs net: %.2f)" % (
        corr.loc["usr", "gpu"] if not np.isnan(corr.loc["usr", "gpu"]) else 0.0,
        corr.loc["gpu", "net_Bps"] if not np.isnan(corr.loc["gpu", "net_Bps"]) else 0.0,
    ))
    corr.to_csv(os.path.join(logdir, "correlation.csv"))
    return perf
