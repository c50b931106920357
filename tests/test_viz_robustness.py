"""Viz server smoke + robustness against corrupt/missing inputs."""

import os
import struct
import threading
import urllib.request

import numpy as np
import pytest

from sofa_amd.config import SofaConfig
from sofa_amd.analyze.main import sofa_analyze
from sofa_amd.preprocess.main import sofa_preprocess
from sofa_amd.preprocess.scs import parse_scs


def test_viz_serves_logdir(tmp_path):
    import http.server
    import socketserver

    logdir = tmp_path
    (logdir / "index.html").write_text("<html>sofa</html>")
    handler = lambda *a, **kw: http.server.SimpleHTTPRequestHandler(  # noqa: E731
        *a, directory=str(logdir), **kw
    )
    with socketserver.TCPServer(("127.0.0.1", 0), handler) as httpd:
        port = httpd.server_address[1]
        t = threading.Thread(target=httpd.serve_forever, daemon=True)
        t.start()
        try:
            body = urllib.request.urlopen(f"http://127.0.0.1:{port}/index.html", timeout=5).read()
            assert b"sofa" in body
        finally:
            httpd.shutdown()


def test_corrupt_scs_rejected(tmp_path):
    bad = tmp_path / "cpusamples.scs"
    bad.write_bytes(b"\x00" * 100)
    with pytest.raises(ValueError):
        parse_scs(str(bad))


def test_truncated_scs_header(tmp_path):
    bad = tmp_path / "cpusamples.scs"
    bad.write_bytes(struct.pack("<I", 0x31534353))  # magic only
    out = parse_scs(str(bad))  # short file -> empty, no crash
    assert len(out.samples) == 0


def test_preprocess_empty_logdir(tmp_path):
    cfg = SofaConfig(logdir=str(tmp_path))
    (tmp_path / "sofa_time.txt").write_text("1000.0\n")
    pre = sofa_preprocess(cfg)
    assert os.path.isfile(os.path.join(str(tmp_path), "report.js"))


def test_analyze_empty_logdir(tmp_path, capsys):
    cfg = SofaConfig(logdir=str(tmp_path))
    sofa_analyze(cfg, {})
    out = capsys.readouterr().out
    assert "Complete!!" in out


def test_analyze_corrupt_csvs(tmp_path, capsys):
    for name in ("cputrace.csv", "gputrace.csv", "vmstat.csv"):
        (tmp_path / name).write_text("not,a,valid\x00csv\n\x01\x02")
    cfg = SofaConfig(logdir=str(tmp_path))
    sofa_analyze(cfg, {})
    assert "Complete!!" in capsys.readouterr().out
