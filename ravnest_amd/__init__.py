"""ravnest_amd — an MI355X-native asynchronous parallel training framework.

Brand-new implementation with the capabilities of ravenprotocol/ravnest
(reference mounted read-only at /root/reference; SURVEY.md maps the parity
surface): offline `clusterize` planning, per-GPU `Node` runtime with the
versioned async-pipeline engine, `Trainer` API, RCCL-over-xGMI comm, DP
parameter averaging, and a hand-written CDNA4 (gfx950) HIP kernel library
for the hot ops (ravnest_amd/ops, ravnest_amd/csrc).
"""

from .strings import NodeTypes, ActionTypes, NodeStatus
from .utils import (set_seed, model_fusion, load_node_json_configs,
                    gpu_usage)
from .planner import clusterize
from .engine import Node, Trainer, ComputeEngine

__version__ = "0.1.0"

__all__ = [
    "Node", "Trainer", "ComputeEngine", "clusterize",
    "set_seed", "model_fusion", "load_node_json_configs", "gpu_usage",
    "NodeTypes", "ActionTypes", "NodeStatus",
]
