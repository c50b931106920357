"""Unified trace schema + SOFATrace record + report.js writer.

Behavioral parity targets:
* SOFATrace record type: reference bin/sofa_models.py:1-7.
* traces_to_json/report.js emission: reference bin/sofa_preprocess.py:343-374
  (each trace series becomes a JS variable with {name,color,data:[{x,y,name}]}
  consumed by sofaboard/timeline.js).
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import List, Optional

import numpy as np
import pandas as pd

from .config import TRACE_COLUMNS


def empty_trace_df() -> pd.DataFrame:
    df = pd.DataFrame(columns=TRACE_COLUMNS)
    return df


def new_trace_df(n: int) -> pd.DataFrame:
    """Pre-sized trace frame with zero/empty defaults."""
    df = pd.DataFrame(
        {
            "timestamp": np.zeros(n, dtype=np.float64),
            "event": np.full(n, -1, dtype=np.float64),
            "duration": np.zeros(n, dtype=np.float64),
            "deviceId": np.full(n, -1, dtype=np.int64),
            "copyKind": np.full(n, -1, dtype=np.int64),
            "payload": np.zeros(n, dtype=np.int64),
            "bandwidth": np.zeros(n, dtype=np.float64),
            "pkt_src": np.zeros(n, dtype=np.int64),
            "pkt_dst": np.zeros(n, dtype=np.int64),
            "pid": np.zeros(n, dtype=np.int64),
            "tid": np.zeros(n, dtype=np.int64),
            "name": np.full(n, "", dtype=object),
            "category": np.zeros(n, dtype=np.int64),
        }
    )
    return df


def trace_df_from(n: int, **cols) -> pd.DataFrame:
    """Single-shot trace-frame construction (no per-column block churn —
    repeated df[col]=... on 1M-row frames costs seconds in pandas)."""
    defaults = {
        "timestamp": lambda: np.zeros(n, dtype=np.float64),
        "event": lambda: np.full(n, -1.0, dtype=np.float64),
        "duration": lambda: np.zeros(n, dtype=np.float64),
        "deviceId": lambda: np.full(n, -1, dtype=np.int64),
        "copyKind": lambda: np.full(n, -1, dtype=np.int64),
        "payload": lambda: np.zeros(n, dtype=np.int64),
        "bandwidth": lambda: np.zeros(n, dtype=np.float64),
        "pkt_src": lambda: np.zeros(n, dtype=np.int64),
        "pkt_dst": lambda: np.zeros(n, dtype=np.int64),
        "pid": lambda: np.zeros(n, dtype=np.int64),
        "tid": lambda: np.zeros(n, dtype=np.int64),
        "name": lambda: np.full(n, "", dtype=object),
        "category": lambda: np.zeros(n, dtype=np.int64),
    }
    data = {}
    for col in TRACE_COLUMNS:
        v = cols.get(col)
        data[col] = defaults[col]() if v is None else v
    return pd.DataFrame(data)


@dataclass
class SOFATrace:
    """One viz series: a DataFrame + display metadata (bin/sofa_models.py:1-7)."""

    name: str = ""
    title: str = ""
    color: str = ""
    x_field: str = "timestamp"
    y_field: str = "duration"
    data: Optional[pd.DataFrame] = None


def downsample(df: pd.DataFrame, ratio: int) -> pd.DataFrame:
    """Keep every ratio-th row (reference list_downsample, bin/sofa_preprocess.py:51-57)."""
    if ratio <= 1 or df is None or len(df) == 0:
        return df
    return df.iloc[::ratio]


def trace_to_js(trace: SOFATrace, plot_ratio: int = 1, max_points: int = 100000) -> str:
    """Serialize one series to a JS assignment for report.js."""
    df = trace.data
    if df is None:
        df = empty_trace_df()
    df = downsample(df, plot_ratio)
    if len(df) > max_points:
        df = df.iloc[:: (len(df) // max_points + 1)]
    points = []
    if len(df) > 0:
        xs = df[trace.x_field].to_numpy(dtype=np.float64, na_value=0.0)
        ys = df[trace.y_field].to_numpy(dtype=np.float64, na_value=0.0)
        names = df["name"].astype(str).to_list() if "name" in df.columns else [""] * len(df)
        for x, y, n in zip(xs, ys, names):
            points.append({"x": round(float(x), 6), "y": float(y), "name": n})
    obj = {
        "name": trace.title or trace.name,
        "color": trace.color or None,
        "turboThreshold": 100000000,
        "data": points,
    }
    return "{} = {};\n".format(trace.name, json.dumps(obj))


def traces_to_json(traces: List[SOFATrace], path: str, plot_ratio: int = 1) -> None:
    """Write report.js: one JS var per series + sofa_traces array.

    Matches the shape sofaboard/timeline.js expects (reference
    bin/sofa_preprocess.py:343-374).
    """
    with open(path, "w") as f:
        names = [t.name for t in traces]
        for t in traces:
            f.write(trace_to_js(t, plot_ratio=plot_ratio))
        f.write("sofa_traces = [{}];\n".format(", ".join(names)))


def write_trace_csv(df: pd.DataFrame, path: str) -> None:
    df.to_csv(
        path,
        index=False,
        columns=TRACE_COLUMNS,
        float_format="%.6f",
    )


def read_trace_csv(path: str) -> pd.DataFrame:
    return pd.read_csv(path)
