"""API request/response models (reference api.py:97-263)."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Union

from pydantic import BaseModel

from ..core.message import Message, MessagePriority, MessageType


class UserCredentials(BaseModel):
    username: str
    password: str


class Token(BaseModel):
    access_token: str
    token_type: str = "bearer"


class MessageRequest(BaseModel):
    """reference api.py:152-160"""

    receiver_id: Optional[str] = None
    content: Union[str, Dict[str, Any], List[Any]]
    message_type: MessageType = MessageType.CHAT
    priority: MessagePriority = MessagePriority.NORMAL
    metadata: Optional[Dict[str, Any]] = None
    visible_to: Optional[List[str]] = None


class MessageResponse(BaseModel):
    """reference api.py:163-193"""

    id: str
    sender_id: str
    receiver_id: Optional[str]
    content: Union[str, Dict[str, Any], List[Any]]
    type: str
    priority: int
    timestamp: float
    status: str
    metadata: Dict[str, Any]
    token_count: Optional[int]
    visible_to: List[str]

    @classmethod
    def from_message(cls, m: Message) -> "MessageResponse":
        return cls(**m.to_dict())


class BroadcastRequest(BaseModel):
    """reference api.py:196-203"""

    content: Union[str, Dict[str, Any], List[Any]]
    message_type: MessageType = MessageType.CHAT
    priority: MessagePriority = MessagePriority.NORMAL
    metadata: Optional[Dict[str, Any]] = None
    exclude_agents: Optional[List[str]] = None


class BroadcastResponse(BaseModel):
    """Declared response for /messages/broadcast. The reference declared
    List[str] but returned a dict (api.py:507-530 — bug SURVEY.md §8.3);
    the dict shape is what its README documents, so that is the model."""

    status: str
    message_id: str


class AgentRegistrationRequest(BaseModel):
    """reference api.py:206-212"""

    agent_id: str
    description: Optional[str] = None
    capabilities: Optional[List[str]] = None
    metadata: Optional[Dict[str, Any]] = None


class AgentGroupRequest(BaseModel):
    """reference api.py:215-219"""

    group_name: str
    agent_ids: List[str]


class GroupMessageRequest(BaseModel):
    """reference api.py:222-229"""

    group_name: str
    content: Union[str, Dict[str, Any], List[Any]]
    message_type: MessageType = MessageType.CHAT
    priority: MessagePriority = MessagePriority.NORMAL
    metadata: Optional[Dict[str, Any]] = None


class GroupMessageResponse(BaseModel):
    """Dict shape (fixes SURVEY.md §8.3 for /groups/message)."""

    status: str
    message_ids: List[str]


class HealthResponse(BaseModel):
    """reference api.py:245-252"""

    status: str
    version: str
    engine: str
    engine_connected: bool
    registered_agents: int
    total_messages: int


class SystemStats(BaseModel):
    """reference api.py:255-263"""

    total_messages: int
    active_agents: int
    messages_by_type: Dict[str, int]
    messages_by_status: Dict[str, int]
    messages_by_agent: Dict[str, Dict[str, int]]
    last_save_time: Optional[float] = None


class AgentLoadResponse(BaseModel):
    agent_id: str
    total_messages: int
    inbox_size: int
    unread_count: int
    processing_rate: float


class LLMDispatchResponse(BaseModel):
    backend_id: str
