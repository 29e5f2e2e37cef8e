"""gRPC prediction client with per-status-code retry/backoff and latency
timing (reference tools/infer_utils.py:14-53, inference.py:33-99)."""
from __future__ import annotations

import functools
import logging
import time
from typing import Callable, Dict, Optional

import grpc
import numpy as np

from . import rpc

log = logging.getLogger("chinesener_amd.serve")

# per-status-code retry budgets (reference tools/infer_utils.py:22-41)
RETRY_BUDGET = {
    grpc.StatusCode.INTERNAL: 1,
    grpc.StatusCode.ABORTED: 3,
    grpc.StatusCode.UNAVAILABLE: 3,
    grpc.StatusCode.DEADLINE_EXCEEDED: 5,
}
MAX_BACKOFF_S = 1.0


def grpc_retry(fn: Callable) -> Callable:
    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        attempts: Dict[grpc.StatusCode, int] = {}
        while True:
            try:
                return fn(*args, **kwargs)
            except grpc.RpcError as e:
                code = e.code()
                budget = RETRY_BUDGET.get(code, 0)
                used = attempts.get(code, 0)
                if used >= budget:
                    raise
                attempts[code] = used + 1
                backoff = min(MAX_BACKOFF_S, 0.05 * (2 ** used))
                log.warning("rpc %s (attempt %d/%d), backing off %.2fs",
                            code.name, used + 1, budget, backoff)
                time.sleep(backoff)
    return wrapped


def timer(fn: Callable) -> Callable:
    """Latency print decorator (reference tools/infer_utils.py:44-53)."""
    @functools.wraps(fn)
    def wrapped(*args, **kwargs):
        t0 = time.perf_counter()
        out = fn(*args, **kwargs)
        log.info("%s took %.1f ms", fn.__name__,
                 (time.perf_counter() - t0) * 1000)
        return out
    return wrapped


class PredictionClient:
    """Channel/stub reuse + typed Predict call (reference InferHelper's
    transport half, inference.py:52-62,88-92)."""

    def __init__(self, host: str = "127.0.0.1", port: int = rpc.DEFAULT_PORT,
                 timeout_s: float = 10.0):
        self.channel = grpc.insecure_channel(f"{host}:{port}")
        self._call = self.channel.unary_unary(
            rpc.FULL_METHOD, request_serializer=None, response_deserializer=None)
        self.timeout_s = timeout_s

    @grpc_retry
    def predict(self, model_name: str, inputs: Dict[str, np.ndarray],
                version: Optional[int] = None) -> Dict:
        req = rpc.dumps(rpc.make_predict_request(model_name, inputs, version))
        raw = self._call(req, timeout=self.timeout_s)
        return rpc.loads(raw)

    def close(self):
        self.channel.close()
