#!/usr/bin/env python3
"""Measure the pure grpcio unary-RPC floor over a unix socket.

Context for profiles/latency_breakdown.md: the Allocate p50 minus this
floor is the plugin's own cost."""

import os
import statistics
import sys
import tempfile
import time
from concurrent import futures

import grpc


def main(n=2000):
    path = os.path.join(tempfile.mkdtemp(), "echo.sock")
    server = grpc.server(futures.ThreadPoolExecutor(4))
    server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler(
            "Echo", {"E": grpc.unary_unary_rpc_method_handler(
                lambda req, ctx: req,
                request_deserializer=lambda b: b,
                response_serializer=lambda b: b)}),))
    server.add_insecure_port("unix:" + path)
    server.start()
    ch = grpc.insecure_channel("unix:" + path)
    call = ch.unary_unary("/Echo/E", request_serializer=lambda b: b,
                          response_deserializer=lambda b: b)
    payload = b"x" * 200  # ≈ an 8-GPU AllocateRequest
    for _ in range(200):
        call(payload)
    ts = []
    for _ in range(n):
        t0 = time.perf_counter()
        call(payload)
        ts.append(time.perf_counter() - t0)
    print("grpcio unary echo floor: p50 %.1f us, p99 %.1f us (n=%d)"
          % (statistics.median(ts) * 1e6,
             sorted(ts)[int(n * 0.99) - 1] * 1e6, n))
    ch.close()
    server.stop(None)


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 2000)
