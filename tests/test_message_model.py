"""Data-model + JSON schema golden tests.

The to_dict/from_dict round trip and the exact wire schema are the
compatibility contract with the reference's intended JSON format
(reference swarmdb/ main.py:91-111, modulo the asdict bug SURVEY.md §8.2).
"""

import json
import uuid

import pytest

from swarmdb_amd import Message, MessagePriority, MessageStatus, MessageType


def test_enums_match_reference_values():
    # reference swarmdb/ main.py:23-51
    assert [t.value for t in MessageType] == [
        "chat", "command", "function_call", "function_result",
        "system", "error", "status",
    ]
    assert [p.value for p in MessagePriority] == [0, 1, 2, 3]
    assert [s.value for s in MessageStatus] == [
        "pending", "delivered", "read", "processed", "failed",
    ]


def test_message_defaults():
    m = Message(sender_id="a", content="hi")
    uuid.UUID(m.id)  # valid uuid4 string
    assert m.receiver_id is None
    assert m.type == MessageType.CHAT
    assert m.priority == MessagePriority.NORMAL
    assert m.status == MessageStatus.PENDING
    assert m.metadata == {}
    assert m.visible_to == []
    assert m.token_count is None
    assert isinstance(m.timestamp, float)


def test_to_dict_golden_schema():
    m = Message(
        id="abc-123",
        sender_id="alice",
        receiver_id="bob",
        content={"k": [1, 2]},
        type=MessageType.COMMAND,
        priority=MessagePriority.HIGH,
        timestamp=1700000000.5,
        status=MessageStatus.DELIVERED,
        metadata={"x": 1},
        token_count=7,
        visible_to=["bob"],
    )
    d = m.to_dict()
    # exact key set and enum flattening (reference swarmdb/ main.py:91-98)
    assert d == {
        "id": "abc-123",
        "sender_id": "alice",
        "receiver_id": "bob",
        "content": {"k": [1, 2]},
        "type": "command",
        "priority": 2,
        "timestamp": 1700000000.5,
        "status": "delivered",
        "metadata": {"x": 1},
        "token_count": 7,
        "visible_to": ["bob"],
    }
    # JSON-serializable as-is (the Kafka wire encode, main.py:466)
    json.dumps(d)


def test_from_dict_round_trip():
    m = Message(
        sender_id="a",
        receiver_id=None,
        content=[1, "two", {"three": 3}],
        type=MessageType.FUNCTION_CALL,
        priority=MessagePriority.CRITICAL,
        status=MessageStatus.READ,
        metadata={"m": True},
        token_count=42,
        visible_to=["b", "c"],
    )
    m2 = Message.from_dict(json.loads(json.dumps(m.to_dict())))
    assert m2 == m


def test_content_types():
    for content in ["plain", {"d": 1}, [1, 2, 3]]:
        m = Message(sender_id="a", content=content)
        assert Message.from_dict(m.to_dict()).content == content


def test_timestamp_none_defaults_to_now():
    m = Message(sender_id="a", content="x", timestamp=None)
    assert m.timestamp > 0
