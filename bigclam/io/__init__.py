from .edgelist import Graph, build_graph, load_graph, parse_edge_array
from .synthetic import (
    planted_partition,
    rmat_edges,
    rmat_graph,
    rmat_graph_with_edges,
    shaped_graph,
)

__all__ = [
    "Graph",
    "build_graph",
    "load_graph",
    "parse_edge_array",
    "planted_partition",
    "rmat_edges",
    "rmat_graph",
    "rmat_graph_with_edges",
    "shaped_graph",
]
