"""sofa diff — swarm-level comparison of two recorded runs.

Parity with reference bin/sofa_ml.py sofa_swarm_diff (:417-539) +
matching_two_dicts_of_swarm (:311-341): read auto_caption.csv from both
logdirs (written by hsg when --enable_swarms was on), greedily fuzzy-match
swarm captions, report per-swarm sample/duration deltas and the caption
intersection rate, write swarm_diff.csv.
"""

from __future__ import annotations

import os

import pandas as pd

from .. import printing as p
from ..config import SofaConfig
from .fuzz import ratio

MATCH_THRESHOLD = 60


def _load_swarms(logdir: str) -> pd.DataFrame:
    path = os.path.join(logdir, "auto_caption.csv")
    if not os.path.isfile(path):
        raise FileNotFoundError(
            f"{path} missing — record/preprocess that logdir with --enable_swarms"
        )
    df = pd.read_csv(path)
    agg = (
        df.groupby("cluster_ID")
        .agg(
            caption=("name", lambda s: s.mode().iloc[0] if len(s) else "?"),
            samples=("duration", "count"),
            duration=("duration", "sum"),
        )
        .reset_index()
    )
    return agg


def sofa_swarm_diff(cfg: SofaConfig) -> pd.DataFrame:
    p.print_title("SOFA diff — swarm comparison")
    base = _load_swarms(cfg.base_logdir)
    match = _load_swarms(cfg.match_logdir)

    used = set()
    rows = []
    matched = 0
    for _, b in base.iterrows():
        best_j, best_score = None, -1
        for j, m in match.iterrows():
            if j in used:
                continue
            s = ratio(str(b["caption"]), str(m["caption"]))
            if s > best_score:
                best_j, best_score = j, s
        if best_j is not None and best_score >= MATCH_THRESHOLD:
            m = match.loc[best_j]
            used.add(best_j)
            matched += 1
            rows.append(
                {
                    "base_cluster": int(b["cluster_ID"]),
                    "match_cluster": int(m["cluster_ID"]),
                    "caption": b["caption"],
                    "match_score": best_score,
                    "base_duration": b["duration"],
                    "match_duration": m["duration"],
                    "duration_delta": m["duration"] - b["duration"],
                    "base_samples": int(b["samples"]),
                    "match_samples": int(m["samples"]),
                }
            )
        else:
            rows.append(
                {
                    "base_cluster": int(b["cluster_ID"]),
                    "match_cluster": -1,
                    "caption": b["caption"],
                    "match_score": best_score if best_j is not None else 0,
                    "base_duration": b["duration"],
                    "match_duration": 0.0,
                    "duration_delta": -b["duration"],
                    "base_samples": int(b["samples"]),
                    "match_samples": 0,
                }
            )
    out = pd.DataFrame(rows)
    intersection_rate = matched / max(len(base), 1)
    print(out.to_string(index=False))
    print("caption intersection rate: %.2f" % intersection_rate)
    # reference wrote into its default sofalog/; falling back to cwd
    # polluted the invoking directory — prefer the match logdir instead
    dest_dir = cfg.logdir if os.path.isdir(cfg.logdir) else cfg.match_logdir
    if not os.path.isdir(dest_dir):
        dest_dir = "."
    dest = os.path.join(dest_dir, "swarm_diff.csv")
    os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
    out.to_csv(dest, index=False)
    p.print_progress(f"wrote {dest}")
    return out
