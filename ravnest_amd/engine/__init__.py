from .compute import ComputeEngine
from .node import Node
from .trainer import Trainer

__all__ = ["ComputeEngine", "Node", "Trainer"]
