"""msgpack wire codec with JSON fallback (reference: utils/msgpack_codec.py:60-80)."""

from __future__ import annotations

import json
from typing import Any, List

try:
    import msgpack

    _HAVE_MSGPACK = True
except Exception:  # pragma: no cover
    msgpack = None
    _HAVE_MSGPACK = False


def encode(obj: Any) -> bytes:
    if _HAVE_MSGPACK:
        return msgpack.packb(obj, use_bin_type=True, default=_default)
    return json.dumps(obj, default=str).encode("utf-8")


def decode(data: bytes) -> Any:
    if _HAVE_MSGPACK:
        return msgpack.unpackb(data, raw=False, strict_map_key=False)
    return json.loads(data.decode("utf-8"))


def _default(obj):
    # Tolerate numpy scalars and other simple objects on the wire.
    for attr in ("item", "tolist"):
        fn = getattr(obj, attr, None)
        if callable(fn):
            try:
                return fn()
            except Exception:
                pass
    return str(obj)


def encode_batch(objs: List[Any]) -> bytes:
    return encode(objs)
