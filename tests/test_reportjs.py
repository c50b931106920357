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
