"""Unit tests for the unified trace schema + report.js writer."""

import json
import os

import numpy as np
import pandas as pd

from sofa_amd.config import TRACE_COLUMNS, Filter, SofaConfig
from sofa_amd.schema import (
    SOFATrace,
    downsample,
    new_trace_df,
    trace_to_js,
    traces_to_json,
    write_trace_csv,
)


def test_trace_columns_match_reference_schema():
    # the 13-column schema (reference bin/sofa_config.py:49-62)
    assert TRACE_COLUMNS == [
        "timestamp", "event", "duration", "deviceId", "copyKind", "payload",
        "bandwidth", "pkt_src", "pkt_dst", "pid", "tid", "name", "category",
    ]


def test_new_trace_df_shape():
    df = new_trace_df(5)
    assert list(df.columns) == TRACE_COLUMNS
    assert len(df) == 5
    assert (df["deviceId"] == -1).all()


def test_downsample():
    df = new_trace_df(100)
    assert len(downsample(df, 10)) == 10
    assert len(downsample(df, 1)) == 100


def test_trace_to_js_roundtrip():
    df = new_trace_df(3)
    df["timestamp"] = [0.1, 0.2, 0.3]
    df["duration"] = [1.0, 2.0, 3.0]
    df["name"] = ["a", "b", "c"]
    t = SOFATrace(name="cpu_traces", title="CPU", color="red", data=df)
    js = trace_to_js(t)
    assert js.startswith("cpu_traces = ")
    obj = json.loads(js.split("= ", 1)[1].rstrip(";\n"))
    assert obj["name"] == "CPU"
    assert len(obj["data"]) == 3
    assert obj["data"][1] == {"x": 0.2, "y": 2.0, "name": "b"}


def test_traces_to_json_writes_array(tmp_path):
    df = new_trace_df(2)
    df["timestamp"] = [0.0, 1.0]
    traces = [
        SOFATrace(name="s1", title="S1", data=df),
        SOFATrace(name="s2", title="S2", data=new_trace_df(0)),
    ]
    out = tmp_path / "report.js"
    traces_to_json(traces, str(out))
    text = out.read_text()
    assert "sofa_traces = [s1, s2];" in text


def test_write_trace_csv_roundtrip(tmp_path):
    df = new_trace_df(4)
    df["timestamp"] = np.arange(4) * 0.5
    df["name"] = ["k1", "k2", "k1", "k3"]
    path = tmp_path / "t.csv"
    write_trace_csv(df, str(path))
    back = pd.read_csv(path)
    assert list(back.columns) == TRACE_COLUMNS
    assert len(back) == 4
    assert back["name"].tolist() == ["k1", "k2", "k1", "k3"]


def test_config_defaults():
    cfg = SofaConfig(logdir="x")
    assert cfg.logdir == "x/"
    assert cfg.cpu_sample_rate == 99      # reference perf -F 99
    assert cfg.sys_mon_rate == 10         # reference sys_mon_rate
    assert cfg.num_iterations == 20       # reference default
    assert cfg.num_swarms == 10
    assert any(f.keyword == "rccl" for f in cfg.gpu_filters)


def test_filter_parsing():
    from sofa_amd.cli import parse_filters

    fs = parse_filters("gemm:red,rccl:blue")
    assert fs[0].keyword == "gemm" and fs[0].color == "red"
    assert fs[1].keyword == "rccl" and fs[1].color == "blue"


def test_trace_df_from_defaults_and_override():
    from sofa_amd.schema import trace_df_from

    df = trace_df_from(3, timestamp=np.array([1.0, 2.0, 3.0]), name=np.array(["a", "b", "c"], dtype=object))
    assert list(df.columns) == TRACE_COLUMNS
    assert df["deviceId"].tolist() == [-1, -1, -1]
    assert df["copyKind"].tolist() == [-1, -1, -1]
    assert df["name"].tolist() == ["a", "b", "c"]
    assert df["timestamp"].tolist() == [1.0, 2.0, 3.0]
