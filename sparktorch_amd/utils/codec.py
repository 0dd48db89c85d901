"""Object <-> string codecs shared by the serialization and persistence layers.

Parity: reference sparktorch/util.py:38-55 (dill+base64 codec) and
sparktorch/pipeline_util.py:35-47,119-130 (dill+zlib comma-decimal codec used by
the pipeline checkpoint format).  Same wire formats, fresh implementation.
"""

from __future__ import annotations

import base64
import zlib
from typing import Any

import dill


def obj_to_bytes(obj: Any) -> bytes:
    return dill.dumps(obj)


def bytes_to_obj(data: bytes) -> Any:
    return dill.loads(data)


def obj_to_b64(obj: Any) -> str:
    """dill -> base64 string (reference util.py:38-44)."""
    return base64.b64encode(dill.dumps(obj)).decode("utf-8")


def b64_to_obj(data: str) -> Any:
    """base64 string -> object (reference util.py:47-55)."""
    return dill.loads(base64.b64decode(data.encode("utf-8")))


# --- pipeline checkpoint byte codec -------------------------------------------------
# The reference smuggles `zlib(dill(obj))` through a JVM StopWordsRemover's
# stopWords as a comma-joined decimal byte string plus a magic GUID
# (pipeline_util.py:27-28,119-130).  These two functions are that exact format.

MAGIC_GUID = "4c1740b00d3c4ff6806a1402321572cb"


def obj_to_stopwords(obj: Any) -> list:
    """Encode obj as the reference's stopWords payload: [csv-bytes, GUID]."""
    compressed = zlib.compress(dill.dumps(obj))
    payload = ",".join(str(b) for b in compressed) + ","
    return [payload, MAGIC_GUID]


def stopwords_to_obj(stop_words: list) -> Any:
    """Decode the [csv-bytes, GUID] payload back to the object."""
    if len(stop_words) < 2 or stop_words[-1] != MAGIC_GUID:
        raise ValueError("not a sparktorch stopWords payload (missing GUID)")
    csv = "".join(stop_words[:-1])
    raw = bytes(int(tok) for tok in csv.split(",") if tok != "")
    return dill.loads(zlib.decompress(raw))


def is_sparktorch_stopwords(stop_words) -> bool:
    try:
        return len(stop_words) >= 2 and stop_words[-1] == MAGIC_GUID
    except TypeError:
        return False
