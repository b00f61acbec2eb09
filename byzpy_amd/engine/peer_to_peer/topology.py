"""Directed communication topology.

Reference parity: engine/peer_to_peer/topology.py:13-38 (complete(n),
ring(n, k) constructors; out/in adjacency).
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Set, Tuple


class Topology:
    def __init__(self, n: int, edges: Iterable[Tuple[int, int]]) -> None:
        self.n = int(n)
        self.out: Dict[int, Set[int]] = {i: set() for i in range(n)}
        self.in_: Dict[int, Set[int]] = {i: set() for i in range(n)}
        for a, b in edges:
            if not (0 <= a < n and 0 <= b < n):
                raise ValueError(f"edge ({a},{b}) out of range")
            if a == b:
                continue
            self.out[a].add(b)
            self.in_[b].add(a)

    def out_neighbors(self, i: int) -> List[int]:
        return sorted(self.out[i])

    def in_neighbors(self, i: int) -> List[int]:
        return sorted(self.in_[i])

    @staticmethod
    def complete(n: int) -> "Topology":
        return Topology(n, [(i, j) for i in range(n) for j in range(n) if i != j])

    @staticmethod
    def ring(n: int, k: int = 1) -> "Topology":
        edges = []
        for i in range(n):
            for step in range(1, k + 1):
                edges.append((i, (i + step) % n))
                edges.append((i, (i - step) % n))
        return Topology(n, edges)
