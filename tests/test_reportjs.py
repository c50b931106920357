"""Validate the report.js artifact structure the sofaboard pages consume."""

import json
import os
import re
import subprocess
import sys

SOFA = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "bin", "sofa")


def test_report_js_structure(tmp_path, native_built):
    logdir = str(tmp_path / "log")
    r = subprocess.run(
        [sys.executable, SOFA, "stat", "dd if=/dev/zero of=/dev/null bs=1M count=8000",
         "--logdir", logdir, "--no_gpu"],
        capture_output=True, text=True, timeout=120,
    )
    assert "Complete!!" in r.stdout
    path = os.path.join(logdir, "report.js")
    text = open(path).read()
    # every assignment is `name = {json};` and the list collects them
    assigns = re.findall(r"^(\w+) = (\{.*\});$", text, re.M)
    assert len(assigns) >= 3  # cpu + mpstat + vmstat at least
    names = []
    for name, payload in assigns:
        obj = json.loads(payload)
        assert "data" in obj and isinstance(obj["data"], list)
        if obj["data"]:
            pt = obj["data"][0]
            assert set(pt) >= {"x", "y"}
        names.append(name)
    m = re.search(r"^sofa_traces = \[(.*)\];$", text, re.M)
    assert m
    listed = [s.strip() for s in m.group(1).split(",")]
    assert set(listed) <= set(names)
    assert "cpu_traces" in listed


def test_chrome_trace_export(tmp_path):
    """chrome_trace.json loads as valid trace-event JSON."""
    import numpy as np

    from sofa_amd.schema import new_trace_df, write_trace_csv
    from sofa_amd.viz.chrome_trace import write_chrome_trace

    df = new_trace_df(4)
    df["timestamp"] = [0.1, 0.2, 0.3, 0.4]
    df["duration"] = 1e-3
    df["deviceId"] = [0, 0, 1, 1]
    df["copyKind"] = [0, 1, 0, 16]
    df["name"] = ["k1", "copy", "k2", "ncclAllReduce"]
    write_trace_csv(df, os.path.join(tmp_path, "gputrace.csv"))
    out = write_chrome_trace(str(tmp_path))
    assert out
    data = json.load(open(out))
    evs = data["traceEvents"]
    assert len(evs) == 4  # 2 kernels + 2 non-kernel rows
    assert all(e["ph"] == "X" and e["dur"] > 0 for e in evs)
    assert {e["pid"] for e in evs} == {"GPU kernels", "GPU copies"}
