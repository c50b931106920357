"""Advisor gRPC server (POTATO-parity transport).

Protocol parity with reference bin/potato_pb2.py:30-208 +
potato_pb2_grpc.py:49-88: unary Greet and Hint methods carrying a hostname +
performance-feature vector, answering with a hint string.  Payloads are JSON
(grpc generic handlers with identity serializers) — the image ships grpcio
but not protoc; the message *content* is the same {name[], value[]} PFV.

Run:  python -m sofa_amd.advisor.server [--port 50051]
"""

from __future__ import annotations

import argparse
import json
from concurrent import futures

import grpc

from .rules import advise, format_hints

SERVICE = "sofa.Advisor"


def _greet(request: bytes, context) -> bytes:
    req = json.loads(request.decode() or "{}")
    return json.dumps({"message": "sofa-amd advisor ready, hello %s" % req.get("name", "")}).encode()


def _hint(request: bytes, context) -> bytes:
    req = json.loads(request.decode() or "{}")
    pfv = req.get("pfv", {})
    names = pfv.get("name", [])
    values = pfv.get("value", [])
    features = {n: float(v) for n, v in zip(names, values)}
    hints = advise(features)
    return json.dumps(
        {
            "hint": format_hints(hints),
            "items": [{"metric": m, "observation": o, "suggestion": s} for m, o, s in hints],
            "docker_image": "",
        }
    ).encode()


def make_server(port: int = 50051) -> grpc.Server:
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
    handlers = {
        "Greet": grpc.unary_unary_rpc_method_handler(_greet),
        "Hint": grpc.unary_unary_rpc_method_handler(_hint),
    }
    server.add_generic_rpc_handlers((grpc.method_handlers_generic_handler(SERVICE, handlers),))
    server.add_insecure_port(f"[::]:{port}")
    return server


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=50051)
    args = ap.parse_args()
    server = make_server(args.port)
    server.start()
    print(f"sofa-amd advisor listening on :{args.port}")
    server.wait_for_termination()


if __name__ == "__main__":
    main()
