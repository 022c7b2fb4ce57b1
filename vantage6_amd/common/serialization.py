"""Algorithm input/result (de)serialization.

Reference parity: ``vantage6-common/vantage6/common/serialization.py`` —
v4 serializes algorithm inputs/results as JSON only (pickle was removed for
security in v4). We keep JSON as the wire format and extend
it (content-prefix tagged, like vantage6 v3's format byte, but explicit and
safe) with numpy/pandas/torch-tensor payloads encoded inside the JSON.
"""
import json
from typing import Any

import numpy as np

from . import bytes_to_base64s, base64s_to_bytes


class SerializationError(Exception):
    pass


def _encode(obj: Any):
    """JSON ``default`` hook: tag ndarray / DataFrame / tensor / bytes."""
    if isinstance(obj, np.ndarray):
        return {
            "__v6amd__": "ndarray",
            "dtype": str(obj.dtype),
            "shape": list(obj.shape),
            "data": bytes_to_base64s(np.ascontiguousarray(obj).tobytes()),
        }
    if isinstance(obj, (np.integer,)):
        return int(obj)
    if isinstance(obj, (np.floating,)):
        return float(obj)
    if isinstance(obj, (np.bool_,)):
        return bool(obj)
    if isinstance(obj, bytes):
        return {"__v6amd__": "bytes", "data": bytes_to_base64s(obj)}
    try:  # pandas is optional at import time
        import pandas as pd

        if isinstance(obj, pd.DataFrame):
            return {"__v6amd__": "dataframe", "data": obj.to_dict(orient="list")}
        if isinstance(obj, pd.Series):
            return {"__v6amd__": "series", "data": obj.tolist(), "name": obj.name}
    except ImportError:
        pass
    try:
        import torch

        if isinstance(obj, torch.Tensor):
            t = obj.detach().cpu()
            # bf16 has no numpy dtype: round-trip through a uint16 view
            if t.dtype == torch.bfloat16:
                arr = t.view(torch.uint16).numpy()
                return {
                    "__v6amd__": "tensor-bf16",
                    "shape": list(t.shape),
                    "data": bytes_to_base64s(np.ascontiguousarray(arr).tobytes()),
                }
            arr = t.numpy()
            return {
                "__v6amd__": "tensor",
                "dtype": str(arr.dtype),
                "shape": list(arr.shape),
                "data": bytes_to_base64s(np.ascontiguousarray(arr).tobytes()),
            }
    except ImportError:
        pass
    raise SerializationError(f"cannot serialize object of type {type(obj)}")


def _decode(d: dict):
    tag = d.get("__v6amd__")
    if tag is None:
        return d
    if tag == "escaped":  # user dict that contained the reserved key
        return {k: v for k, v in d["items"]}
    if tag == "ndarray":
        arr = np.frombuffer(base64s_to_bytes(d["data"]), dtype=np.dtype(d["dtype"]))
        return arr.reshape(d["shape"]).copy()
    if tag == "bytes":
        return base64s_to_bytes(d["data"])
    if tag == "dataframe":
        import pandas as pd

        return pd.DataFrame(d["data"])
    if tag == "series":
        import pandas as pd

        return pd.Series(d["data"], name=d.get("name"))
    if tag == "tensor":
        import torch

        arr = np.frombuffer(base64s_to_bytes(d["data"]), dtype=np.dtype(d["dtype"]))
        return torch.from_numpy(arr.reshape(d["shape"]).copy())
    if tag == "tensor-bf16":
        import torch

        arr = np.frombuffer(base64s_to_bytes(d["data"]), dtype=np.uint16)
        t = torch.from_numpy(arr.copy()).view(torch.bfloat16)
        return t.reshape(d["shape"])
    raise SerializationError(f"unknown serialization tag {tag!r}")


def _escape(o: Any) -> Any:
    """User dicts that happen to contain the reserved tag key would be
    mis-decoded as tagged payloads — wrap them as an 'escaped' tag whose
    items ride as [key, value] PAIRS (a pairs list cannot collide with the
    object_hook, which fires on every dict bottom-up)."""
    if isinstance(o, dict):
        walked = {k: _escape(v) for k, v in o.items()}
        if "__v6amd__" in walked:
            return {"__v6amd__": "escaped",
                    "items": [[k, v] for k, v in walked.items()]}
        return walked
    if isinstance(o, (list, tuple)):
        return [_escape(v) for v in o]
    return o


def serialize(data: Any) -> bytes:
    """Serialize an algorithm input/result to bytes (JSON, UTF-8)."""
    return json.dumps(_escape(data), default=_encode).encode("utf-8")


def deserialize(blob: bytes) -> Any:
    """Inverse of :func:`serialize`."""
    return json.loads(blob.decode("utf-8"), object_hook=_decode)
