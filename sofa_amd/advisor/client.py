"""Advisor client (reference get_hint, bin/sofa_analyze.py:49-73)."""

from __future__ import annotations

import json
import socket
from typing import Optional

import pandas as pd

from .rules import advise, format_hints

SERVICE = "sofa.Advisor"


def local_hints(features_df: pd.DataFrame) -> str:
    features = dict(zip(features_df["name"], features_df["value"]))
    return format_hints(advise(features))


def get_hint(server: str, features_df: pd.DataFrame, timeout: float = 5.0) -> Optional[str]:
    """Query the advisor; `server` is host[:port] or 'local'."""
    if server in ("local", "localhost-rules"):
        return local_hints(features_df)
    import grpc

    if ":" not in server:
        server = server + ":50051"
    pfv = {
        "name": [str(n) for n in features_df["name"]],
        "value": [float(v) for v in features_df["value"]],
    }
    payload = json.dumps({"hostname": socket.gethostname(), "pfv": pfv}).encode()
    channel = grpc.insecure_channel(server)
    try:
        stub = channel.unary_unary(f"/{SERVICE}/Hint")
        resp = stub(payload, timeout=timeout)
        return json.loads(resp.decode()).get("hint")
    finally:
        channel.close()
