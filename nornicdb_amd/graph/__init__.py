"""Graph analytics: CSR adjacency + HIP/CPU algorithms."""

from .csr import CSRGraph, from_edges, from_engine, random_graph
from .fastrp import fastrp_embeddings
from .algos import (astar, betweenness_centrality, bfs_distances,
                    closeness_centrality, clustering_coefficient,
                    connected_components, degree_centrality, dijkstra,
                    label_propagation, louvain, modularity, pagerank,
                    shortest_path, triangle_count)

__all__ = ["CSRGraph", "from_edges", "from_engine", "random_graph",
           "pagerank", "bfs_distances", "dijkstra", "astar", "shortest_path",
           "connected_components", "label_propagation", "degree_centrality",
           "closeness_centrality", "betweenness_centrality", "triangle_count",
           "clustering_coefficient", "modularity", "louvain", "fastrp_embeddings"]
