"""Wire protocol for the serving path.

The reference speaks TF-Serving's PredictionService gRPC with a
serialized tf.train.Example (inference.py:55-62). protoc is not in this
image, so the rebuild defines the same unary Predict RPC with gRPC
generic handlers and msgpack framing: request = {model_spec{name,
version, signature}, inputs{key: ndarray}}, response = {outputs{key:
ndarray}} — structurally the same contract, no compiled stubs needed."""
from __future__ import annotations

from typing import Any, Dict

import msgpack
import numpy as np

SERVICE = "chinesener.PredictionService"
METHOD = "Predict"
FULL_METHOD = f"/{SERVICE}/{METHOD}"
DEFAULT_PORT = 8500  # reference server.sh exposes 8500


def _pack_nd(a: np.ndarray) -> Dict[str, Any]:
    a = np.ascontiguousarray(a)
    return {"__nd__": True, "dtype": a.dtype.str, "shape": list(a.shape),
            "data": a.tobytes()}


def _default(obj):
    if isinstance(obj, np.ndarray):
        return _pack_nd(obj)
    if isinstance(obj, (np.integer,)):
        return int(obj)
    if isinstance(obj, (np.floating,)):
        return float(obj)
    raise TypeError(f"unpackable type {type(obj)}")


def _hook(obj):
    if isinstance(obj, dict) and obj.get("__nd__"):
        return np.frombuffer(obj["data"], dtype=np.dtype(obj["dtype"])) \
            .reshape(obj["shape"]).copy()
    return obj


def dumps(msg: Dict) -> bytes:
    return msgpack.packb(msg, default=_default, use_bin_type=True)


def loads(raw: bytes) -> Dict:
    return msgpack.unpackb(raw, object_hook=_hook, raw=False,
                           strict_map_key=False)


def make_predict_request(model_name: str, inputs: Dict[str, np.ndarray],
                         version: int | None = None,
                         signature: str = "serving_default") -> Dict:
    return {"model_spec": {"name": model_name, "version": version,
                           "signature_name": signature},
            "inputs": inputs}
