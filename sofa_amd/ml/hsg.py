"""HSG — hierarchical swarm grouping of CPU samples.

Parity with reference bin/sofa_ml.py hsg_v2 (:243-287) + swarms_to_sofatrace
(:289-309): cluster samples by event (= log10 of the sampled IP) with
average-linkage agglomerative clustering into cfg.num_swarms swarms, caption
each swarm with the modal resolved function name, emit swarm series for the
timeline, swarms_report.txt and auto_caption.csv (the input to sofa diff).
"""

from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np
import pandas as pd

from ..schema import SOFATrace

SWARM_COLORS = [
    "#e6194b", "#3cb44b", "#ffe119", "#4363d8", "#f58231", "#911eb4",
    "#46f0f0", "#f032e6", "#bcf60c", "#fabebe", "#008080", "#e6beff",
]


def _caption(names: pd.Series) -> str:
    if len(names) == 0:
        return "?"
    return names.mode().iloc[0]


def hsg_cluster(
    df_cpu: pd.DataFrame, num_swarms: int, logdir: str = ""
) -> Tuple[pd.DataFrame, List[str]]:
    """Assign cluster_ID to every sample; returns (df + cluster_ID, captions)."""
    df = df_cpu.copy()
    n = len(df)
    if n == 0:
        return df.assign(cluster_ID=[]), []
    k = min(num_swarms, n)
    X = df[["event"]].to_numpy(dtype=np.float64)
    if n > 20000:
        # agglomerative is O(n^2): cluster a sample, assign the rest by
        # nearest centroid (the reference just ate the memory; SURVEY.md §3.2
        # flags it as a hot spot)
        idx = np.random.default_rng(0).choice(n, 20000, replace=False)
        from sklearn.cluster import AgglomerativeClustering

        sub_labels = AgglomerativeClustering(n_clusters=k, linkage="average").fit_predict(X[idx])
        cents = np.array([X[idx][sub_labels == c].mean() for c in range(k)])  # (k,)
        labels = np.argmin(np.abs(X - cents.reshape(1, -1)), axis=1)
    else:
        from sklearn.cluster import AgglomerativeClustering

        labels = AgglomerativeClustering(n_clusters=k, linkage="average").fit_predict(X)
    df["cluster_ID"] = labels

    captions = []
    for c in range(k):
        sel = df[df["cluster_ID"] == c]
        captions.append(_caption(sel["name"].astype(str)))

    if logdir:
        with open(os.path.join(logdir, "swarms_report.txt"), "w") as f:
            f.write("swarm report: %d swarms over %d samples\n" % (k, n))
            for c in range(k):
                sel = df[df["cluster_ID"] == c]
                f.write(
                    "swarm %2d: %6d samples, %9.4f s total -- %s\n"
                    % (c, len(sel), sel["duration"].sum(), captions[c])
                )
        cap = df[["timestamp", "event", "duration", "deviceId", "pid", "tid", "name", "cluster_ID"]]
        cap.to_csv(os.path.join(logdir, "auto_caption.csv"), index=False)
    return df, captions


def swarms_to_traces(df: pd.DataFrame, captions: List[str], logdir: str = "") -> List[SOFATrace]:
    out = []
    if "cluster_ID" not in df.columns:
        return out
    for c, caption in enumerate(captions):
        sel = df[df["cluster_ID"] == c]
        if len(sel) == 0:
            continue
        short = caption if len(caption) < 60 else caption[:57] + "..."
        out.append(
            SOFATrace(
                name=f"swarm_{c}",
                title=f"swarm{c}: {short}",
                color=SWARM_COLORS[c % len(SWARM_COLORS)],
                data=sel,
            )
        )
    return out
