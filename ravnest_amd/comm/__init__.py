from .backend import CommBackend, Edge
from .p2p import Channel, Message, ACTION_CODES, CODE_ACTIONS
from .collectives import average_parameters, average_optimizer_state

__all__ = [
    "CommBackend", "Edge", "Channel", "Message",
    "ACTION_CODES", "CODE_ACTIONS",
    "average_parameters", "average_optimizer_state",
]
