"""Message data model and JSON wire/disk schema.

MI355X-native rebuild of the reference data model (reference:
``swarmdb/ main.py:23-127``). The JSON formats emitted here are the
compatibility contract: ``Message.to_dict()`` must produce the exact dict
shape the reference *intends* (all fields, enums flattened to ``.value`` —
reference ``swarmdb/ main.py:91-98``; the reference's own implementation is
broken, see SURVEY.md §8.2, we implement the intended schema).

History-file format (reference ``swarmdb/ main.py:877-886``)::

    {"messages": {msg_id: msg_dict}, "agent_inbox": {agent_id: [msg_id]},
     "registered_agents": [agent_id], "timestamp": float,
     "message_count": int}

Archive format for flushed messages (reference ``swarmdb/ main.py:1184-1196``):
a bare ``{msg_id: msg_dict}`` object.
"""

from __future__ import annotations

import time
import uuid
from enum import Enum
from typing import Any, Dict, List, Optional, Union

from pydantic import BaseModel, Field, field_validator


class MessageType(str, Enum):
    """Message kinds (reference swarmdb/ main.py:23-32)."""

    CHAT = "chat"
    COMMAND = "command"
    FUNCTION_CALL = "function_call"
    FUNCTION_RESULT = "function_result"
    SYSTEM = "system"
    ERROR = "error"
    STATUS = "status"


class MessagePriority(int, Enum):
    """Priority levels (reference swarmdb/ main.py:35-41)."""

    LOW = 0
    NORMAL = 1
    HIGH = 2
    CRITICAL = 3


class MessageStatus(str, Enum):
    """Delivery lifecycle (reference swarmdb/ main.py:44-51)."""

    PENDING = "pending"
    DELIVERED = "delivered"
    READ = "read"
    PROCESSED = "processed"
    FAILED = "failed"


class Message(BaseModel):
    """One message record (reference swarmdb/ main.py:54-111).

    ``receiver_id is None`` means broadcast. ``visible_to`` empty means
    visible to everyone the routing rules allow.
    """

    id: str = Field(default_factory=lambda: str(uuid.uuid4()))
    sender_id: str
    receiver_id: Optional[str] = None
    content: Union[str, Dict[str, Any], List[Any]]
    type: MessageType = MessageType.CHAT
    priority: MessagePriority = MessagePriority.NORMAL
    timestamp: float = Field(default_factory=time.time)
    status: MessageStatus = MessageStatus.PENDING
    metadata: Dict[str, Any] = Field(default_factory=dict)
    token_count: Optional[int] = None
    visible_to: List[str] = Field(default_factory=list)

    @field_validator("timestamp", mode="before")
    @classmethod
    def _default_timestamp(cls, v: Any) -> float:
        if v is None:
            return time.time()
        return float(v)

    def to_dict(self) -> Dict[str, Any]:
        """The canonical JSON dict: every field, enums as their values.

        This is the exact wire format produced for transport and
        persistence (reference intent at swarmdb/ main.py:91-98; the
        reference's dataclasses.asdict call is broken — SURVEY.md §8.2).
        """
        return {
            "id": self.id,
            "sender_id": self.sender_id,
            "receiver_id": self.receiver_id,
            "content": self.content,
            "type": self.type.value,
            "priority": self.priority.value,
            "timestamp": self.timestamp,
            "status": self.status.value,
            "metadata": self.metadata,
            "token_count": self.token_count,
            "visible_to": self.visible_to,
        }

    @classmethod
    def from_dict(cls, data: Dict[str, Any], validate: bool = True) -> "Message":
        """Inverse of :meth:`to_dict` (reference swarmdb/ main.py:101-111).
        ``validate=False`` skips pydantic validation for trusted input
        (our own history files on the bulk load path)."""
        d = dict(data)
        if "type" in d:
            d["type"] = MessageType(d["type"])
        if "priority" in d:
            d["priority"] = MessagePriority(d["priority"])
        if "status" in d:
            d["status"] = MessageStatus(d["status"])
        if not validate:
            d.setdefault("metadata", {})
            d.setdefault("visible_to", [])
            d.setdefault("token_count", None)
            d.setdefault("receiver_id", None)
            return cls.model_construct(**d)
        return cls(**d)
