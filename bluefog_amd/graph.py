# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Minimal weighted directed-graph container.

The reference framework exposes its virtual topologies as ``networkx.DiGraph``
objects built from adjacency matrices (reference: bluefog/common/
topology_util.py). networkx is not a dependency of this framework; this
module provides the small surface the topology code and the public API
actually use, backed directly by the adjacency matrix A where ``A[i, j]`` is
the weight of edge i -> j.

A real ``networkx.DiGraph`` is accepted anywhere a :class:`DiGraph` is, via
:func:`as_digraph` duck-typing conversion.
"""

from typing import Iterable, List, Optional

import numpy as np


class DiGraph:
    """Weighted directed graph over nodes ``0..n-1`` backed by an adjacency
    matrix. Edge i->j exists iff ``A[i, j] != 0``."""

    def __init__(self, adjacency: np.ndarray):
        adjacency = np.asarray(adjacency, dtype=np.float64)
        if adjacency.ndim != 2 or adjacency.shape[0] != adjacency.shape[1]:
            raise ValueError("adjacency must be a square matrix")
        self._A = adjacency

    # -- construction ------------------------------------------------------
    @classmethod
    def from_numpy_array(cls, A: np.ndarray) -> "DiGraph":
        return cls(A)

    # -- basic queries -----------------------------------------------------
    def number_of_nodes(self) -> int:
        return self._A.shape[0]

    def number_of_edges(self) -> int:
        return int(np.count_nonzero(self._A))

    def nodes(self) -> Iterable[int]:
        return range(self.number_of_nodes())

    def has_edge(self, u: int, v: int) -> bool:
        return self._A[u, v] != 0

    def __getitem__(self, u: int):
        """Successor-weight mapping, networkx-style ``G[u][v]['weight']``."""
        return {v: {"weight": float(self._A[u, v])} for v in self.successors(u)}

    def predecessors(self, v: int) -> List[int]:
        """Ranks u with an edge u -> v, ascending (deterministic, unlike
        networkx insertion order; the reference sorts them before use anyway,
        basics.py:355-360)."""
        return [int(u) for u in np.nonzero(self._A[:, v])[0]]

    def successors(self, u: int) -> List[int]:
        return [int(v) for v in np.nonzero(self._A[u, :])[0]]

    def in_degree(self, v: int) -> int:
        return len(self.predecessors(v))

    def out_degree(self, u: int) -> int:
        return len(self.successors(u))

    def degree(self, v: int) -> int:
        return self.in_degree(v) + self.out_degree(v)

    def to_numpy_array(self) -> np.ndarray:
        return self._A.copy()

    def copy(self) -> "DiGraph":
        return DiGraph(self._A.copy())

    def __eq__(self, other) -> bool:
        if not isinstance(other, DiGraph):
            return NotImplemented
        return self._A.shape == other._A.shape and bool((self._A == other._A).all())

    def __repr__(self) -> str:  # pragma: no cover
        return f"DiGraph(n={self.number_of_nodes()}, edges={self.number_of_edges()})"


def as_digraph(topology) -> Optional[DiGraph]:
    """Accept our DiGraph or any networkx-like digraph (duck-typed through its
    adjacency matrix) and return a :class:`DiGraph`."""
    if topology is None:
        return None
    if isinstance(topology, DiGraph):
        return topology
    # networkx duck-typing: rebuild from the adjacency matrix.
    try:
        import networkx as nx  # type: ignore

        if isinstance(topology, nx.DiGraph):
            return DiGraph(nx.to_numpy_array(topology, nodelist=sorted(topology.nodes())))
    except ImportError:
        pass
    if hasattr(topology, "to_numpy_array"):
        return DiGraph(topology.to_numpy_array())
    raise TypeError(
        f"topology must be a bluefog_amd DiGraph (or networkx.DiGraph); got {type(topology)}"
    )
