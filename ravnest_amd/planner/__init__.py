from .splitter import split_model_by_proportions, trace_model
from .placement import NodeSpec, Cluster, form_clusters
from .clusterize import clusterize

__all__ = [
    "clusterize",
    "split_model_by_proportions",
    "trace_model",
    "NodeSpec",
    "Cluster",
    "form_clusters",
]
