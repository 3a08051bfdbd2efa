"""swarmdb_amd — MI355X-native agent message queue + LLM load balancer.

A from-scratch rebuild of The-Swarm-Corporation/SwarmDB's capabilities
(agent registration, point-to-point / broadcast / group messaging, JSON
persistence, JWT-authenticated REST API, LLM-backend load balancing) with
the Kafka tier replaced by a GPU-resident MPMC ring buffer in MI355X HBM3E:
hand-written CDNA4 HIP kernels for enqueue / dequeue / fan-out /
priority-sort / least-loaded reduction, RCCL all-to-all over xGMI for
cross-GPU routing, and pinned hipMemcpyAsync side-stream spill for history
persistence.
"""

from .core.config import QueueConfig
from .core.message import Message, MessagePriority, MessageStatus, MessageType
from .runtime.facade import SwarmsDB

__version__ = "0.2.0"

__all__ = [
    "SwarmsDB",
    "QueueConfig",
    "Message",
    "MessageType",
    "MessagePriority",
    "MessageStatus",
]
