"""Payload wire encoding + derived message ids.

Slot payload layout (shared with the HIP kernels, csrc/swarmq_common.h)::

    [content bytes (content_len, from the record header)]
    [extras JSON (payload_len - content_len), present iff FLAG_HAS_EXTRAS]

``content`` is raw utf-8 for str content, JSON for dict/list content
(FLAG_JSON_CONTENT). ``extras`` carries the fields that have no binary
header slot: ``{"id": ..., "metadata": {...}, "visible_to": [...]}``.

Batch-path messages skip extras entirely and derive their id from
(rank, seq) as a UUID-shaped string (version nibble 8 so it can never
collide with the compat path's uuid4 ids).
"""

from __future__ import annotations

import json
import re
from typing import Any, Optional, Tuple

_DERIVED_RE = re.compile(
    r"^([0-9a-f]{8})-0000-8000-8000-([0-9a-f]{12})$"
)


def derived_id(rank: int, seq: int) -> str:
    """UUID-shaped id carrying (rank, seq); parseable, zero host state."""
    return "%08x-0000-8000-8000-%012x" % (rank & 0xFFFFFFFF, seq & 0xFFFFFFFFFFFF)


def parse_derived_id(msg_id: str) -> Optional[Tuple[int, int]]:
    m = _DERIVED_RE.match(msg_id)
    if not m:
        return None
    return int(m.group(1), 16), int(m.group(2), 16)


def encode_content(content: Any) -> Tuple[bytes, bool]:
    """Returns (content_bytes, is_json)."""
    if isinstance(content, str):
        return content.encode("utf-8"), False
    return json.dumps(content).encode("utf-8"), True


def decode_content(data: bytes, is_json: bool) -> Any:
    if is_json:
        return json.loads(data.decode("utf-8"))
    return data.decode("utf-8")


def encode_extras(msg_id: Optional[str], metadata: dict, visible_to: list) -> bytes:
    ex: dict = {}
    if msg_id is not None:
        ex["id"] = msg_id
    if metadata:
        ex["metadata"] = metadata
    if visible_to:
        ex["visible_to"] = visible_to
    if not ex:
        return b""
    return json.dumps(ex).encode("utf-8")


def decode_extras(data: bytes) -> dict:
    if not data:
        return {}
    return json.loads(data.decode("utf-8"))
