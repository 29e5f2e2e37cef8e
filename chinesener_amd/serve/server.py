"""gRPC prediction server (the reference's TF-Serving docker role,
server.sh:1-5): loads exported models from ./serving_model/{name}/{ver}/,
captures the hipGraph inference engine, replays recorded warmup requests
at startup, then serves unary Predict RPCs on port 8500."""
from __future__ import annotations

import argparse
import logging
import os
import pickle
import sys
import time
from concurrent import futures
from typing import Dict

import grpc

from ..config import EXPORT_DIR
from . import rpc
from .engine import InferenceEngine, MicroBatcher
from .export import latest_version_dir

log = logging.getLogger("chinesener_amd.serve")

WARMUP_FILE = os.path.join("assets.extra", "serving_warmup_requests")


def load_warmup(model_dir: str):
    path = os.path.join(model_dir, WARMUP_FILE)
    if not os.path.exists(path):
        return []
    with open(path, "rb") as f:
        return pickle.load(f)


class PredictionServicer:
    def __init__(self, engines: Dict[str, InferenceEngine]):
        self.engines = engines

    def predict(self, request_bytes: bytes, context) -> bytes:
        try:
            req = rpc.loads(request_bytes)
            name = req["model_spec"]["name"]
        except Exception as e:
            context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                          f"malformed request: {e}")
        engine = self.engines.get(name)
        if engine is None:
            # context.abort raises; no code runs after it
            context.abort(grpc.StatusCode.NOT_FOUND,
                          f"model '{name}' not loaded "
                          f"(loaded: {sorted(self.engines)})")
        try:
            t0 = time.perf_counter()
            pred = engine.predict(req["inputs"])
            ms = (time.perf_counter() - t0) * 1000
            return rpc.dumps({"outputs": {"pred_ids": pred},
                              "model_spec": req["model_spec"],
                              "latency_ms": ms})
        except Exception as e:  # surface as INTERNAL with the message
            log.exception("predict failed")
            context.abort(grpc.StatusCode.INTERNAL, str(e))


def build_server(engines: Dict[str, InferenceEngine], port: int,
                 max_workers: int = 4) -> grpc.Server:
    servicer = PredictionServicer(engines)
    handler = grpc.method_handlers_generic_handler(
        rpc.SERVICE,
        {rpc.METHOD: grpc.unary_unary_rpc_method_handler(
            servicer.predict,
            request_deserializer=None, response_serializer=None)})
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=max_workers))
    server.add_generic_rpc_handlers((handler,))
    server.add_insecure_port(f"[::]:{port}")
    return server


def serve(model_names, export_root: str = EXPORT_DIR,
          port: int = rpc.DEFAULT_PORT, wait: bool = True,
          use_graph=None, batch_sizes=(1, 4, 8), max_workers: int = 4,
          batching: bool = True, window_ms: float = 0.3):
    engines = {}
    for name in model_names:
        eng = InferenceEngine(name, export_root, batch_sizes=batch_sizes,
                              use_graph=use_graph)
        warm = load_warmup(latest_version_dir(name, export_root))
        eng.warmup(warm)
        log.info("loaded %s (%d warmup requests replayed)", name, len(warm))
        # micro-batching: coalesce concurrent clients into one replay
        # (the engine itself serializes under a lock; see MicroBatcher)
        engines[name] = MicroBatcher(eng, window_ms=window_ms) if batching else eng
    server = build_server(engines, port, max_workers=max_workers)
    server.start()
    log.info("serving %s on :%d", sorted(engines), port)
    if wait:
        server.wait_for_termination()
    return server


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", required=True,
                    help="comma-separated exported model names")
    ap.add_argument("--export_root", default=EXPORT_DIR)
    ap.add_argument("--port", type=int, default=rpc.DEFAULT_PORT)
    ap.add_argument("--no_graph", action="store_true",
                    help="disable hipGraph capture (debug)")
    ap.add_argument("--no_batching", action="store_true",
                    help="disable concurrent-request micro-batching")
    args = ap.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    serve(args.model.split(","), args.export_root, args.port,
          use_graph=False if args.no_graph else None,
          batching=not args.no_batching)
    return 0


if __name__ == "__main__":
    sys.exit(main())
