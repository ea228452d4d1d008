"""Role / action / status enums.

Capability parity with the reference's ravnest/strings.py:1-24
(NodeTypes, ActionTypes, BufferStatus, NodeStatus), re-expressed for the
MI355X runtime where buffers are device-resident queues rather than gRPC
shared-memory slots.
"""
from enum import Enum


class NodeTypes(str, Enum):
    ROOT = "root"
    STEM = "stem"
    LEAF = "leaf"

    def __str__(self):  # pragma: no cover - trivial
        return self.value


class ActionTypes(str, Enum):
    ROOT_FORWARD = "root_forward"
    FORWARD = "forward"
    BACKWARD = "backward"
    FIND_LOSS = "find_loss"
    NO_GRAD_FORWARD = "no_grad_forward"
    ACCURACY = "accuracy"
    VAL_ACCURACY = "val_accuracy"
    PREDICTION = "prediction"
    SAVE_SUBMODEL = "save_submodel"
    STOP = "stop"  # extra vs reference: clean pipeline shutdown cascade

    def __str__(self):  # pragma: no cover - trivial
        return self.value


class NodeStatus(str, Enum):
    IDLE = "idle"
    FORWARD = "forward"
    BACKWARD = "backward"

    def __str__(self):  # pragma: no cover - trivial
        return self.value
