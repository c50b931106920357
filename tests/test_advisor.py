"""Advisor tests: rule engine + gRPC round trip over localhost."""

import pandas as pd
import pytest

from sofa_amd.advisor.rules import advise, format_hints
from sofa_amd.advisor.client import get_hint, local_hints


def test_rules_comm_bound():
    f = {"iter_step_time": 0.1, "iter_coll_time": 0.05, "iter_copy_time": 0.0}
    hints = advise(f)
    metrics = [h[0] for h in hints]
    assert "iter_coll_time" in metrics
    text = format_hints(hints)
    assert "bucket_cap_mb" in text


def test_rules_iow_bound():
    f = {"dominant_iow_ratio": 0.5}
    assert any(h[0] == "dominant_iow_ratio" for h in advise(f))


def test_rules_no_bottleneck():
    hints = advise({})
    assert hints[0][0] == "overall"


def test_local_hints_df():
    df = pd.DataFrame({"name": ["dominant_iow_ratio"], "value": [0.9]})
    assert "NVMe" in local_hints(df)


def test_grpc_round_trip():
    from sofa_amd.advisor.server import make_server

    server = make_server(port=50911)
    server.start()
    try:
        df = pd.DataFrame(
            {"name": ["iter_step_time", "iter_coll_time"], "value": [0.1, 0.05]}
        )
        hint = get_hint("127.0.0.1:50911", df, timeout=10)
        assert hint and "bucket_cap_mb" in hint
    finally:
        server.stop(0)
