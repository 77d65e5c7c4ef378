"""BallTree / ConditionalBallTree — exact maximum-inner-product search.

Public-class parity with core/.../nn/BallTree.scala:109 (makeBallTree split,
upperBoundMaximumInnerProduct:53, bound-pruned traverseTree:122-143),
ConditionalBallTree.scala:202 (label-filtered search with per-node label
sets) and BoundedPriorityQueue.  The KNN estimators use the brute-force MFMA
path (models/knn.py) for throughput; these trees serve the single-query /
CPU-side use the reference exposes directly."""
from __future__ import annotations

import heapq
from typing import List, Optional, Sequence, Set

import numpy as np


class BoundedPriorityQueue:
    """Keep the k best (largest) items (nn/BoundedPriorityQueue.scala)."""

    def __init__(self, k: int):
        self.k = k
        self._heap: List = []  # min-heap of (score, counter, item)
        self._n = 0

    def offer(self, score: float, item):
        if len(self._heap) < self.k:
            heapq.heappush(self._heap, (score, self._n, item))
        elif score > self._heap[0][0]:
            heapq.heapreplace(self._heap, (score, self._n, item))
        self._n += 1

    @property
    def min_score(self) -> float:
        return self._heap[0][0] if len(self._heap) == self.k else -np.inf

    def items(self):
        return [(s, it) for s, _, it in sorted(self._heap, reverse=True)]


class _Node:
    __slots__ = ("center", "radius", "left", "right", "idx", "labels")

    def __init__(self, center, radius, left=None, right=None, idx=None,
                 labels=None):
        self.center = center
        self.radius = radius
        self.left = left
        self.right = right
        self.idx = idx          # leaf point indices
        self.labels = labels    # label set (conditional tree)


class BallTree:
    def __init__(self, points: np.ndarray, values: Optional[Sequence] = None,
                 leaf_size: int = 50):
        self.points = np.asarray(points, dtype=np.float32)
        self.values = list(values) if values is not None else list(
            range(len(self.points)))
        self.leaf_size = leaf_size
        self.root = self._build(np.arange(len(self.points)))

    def _make_node(self, idx: np.ndarray) -> _Node:
        pts = self.points[idx]
        center = pts.mean(axis=0)
        radius = float(np.sqrt(((pts - center) ** 2).sum(axis=1).max()))
        return _Node(center, radius)

    def _build(self, idx: np.ndarray) -> _Node:
        node = self._make_node(idx)
        if len(idx) <= self.leaf_size:
            node.idx = idx
            return node
        pts = self.points[idx]
        # split along the direction between the two approximately-farthest
        # points (reference's makeBallTree split rule)
        a = pts[int(((pts - pts[0]) ** 2).sum(axis=1).argmax())]
        b = pts[int(((pts - a) ** 2).sum(axis=1).argmax())]
        d = a - b
        proj = pts @ d
        order = np.argsort(proj, kind="stable")
        half = len(idx) // 2
        node.left = self._build(idx[order[:half]])
        node.right = self._build(idx[order[half:]])
        return node

    @staticmethod
    def _upper_bound_mip(q: np.ndarray, qn: float, node: _Node) -> float:
        """Max possible q·x for x in the ball (upperBoundMaximumInnerProduct)."""
        return float(q @ node.center) + qn * node.radius

    def find_maximum_inner_products(self, query: np.ndarray, k: int = 1):
        """Top-k (value, inner_product) by bound-pruned DFS."""
        q = np.asarray(query, dtype=np.float32)
        qn = float(np.linalg.norm(q))
        best = BoundedPriorityQueue(k)

        def dfs(node: _Node):
            if self._upper_bound_mip(q, qn, node) <= best.min_score:
                return
            if node.idx is not None:
                ips = self.points[node.idx] @ q
                for i, ip in zip(node.idx, ips):
                    best.offer(float(ip), int(i))
                return
            lb = self._upper_bound_mip(q, qn, node.left)
            rb = self._upper_bound_mip(q, qn, node.right)
            first, second = ((node.left, node.right) if lb >= rb
                             else (node.right, node.left))
            dfs(first)
            dfs(second)

        dfs(self.root)
        return [(self.values[i], s) for s, i in best.items()]


class ConditionalBallTree(BallTree):
    """Label-filtered MIPS (ConditionalBallTree.scala:202): each point has a
    label; queries restrict matches to a conditioner label set, pruning
    subtrees whose label set misses the conditioner (ReverseIndex analog)."""

    def __init__(self, points, labels: Sequence, values=None, leaf_size=50):
        self.point_labels = list(labels)
        super().__init__(points, values, leaf_size)
        self._annotate(self.root)

    def _annotate(self, node: _Node) -> Set:
        if node.idx is not None:
            node.labels = {self.point_labels[i] for i in node.idx}
        else:
            node.labels = self._annotate(node.left) | self._annotate(node.right)
        return node.labels

    def find_maximum_inner_products(self, query, conditioner: Set, k: int = 1):
        q = np.asarray(query, dtype=np.float32)
        qn = float(np.linalg.norm(q))
        cond = set(conditioner)
        best = BoundedPriorityQueue(k)

        def dfs(node: _Node):
            if not (node.labels & cond):
                return
            if self._upper_bound_mip(q, qn, node) <= best.min_score:
                return
            if node.idx is not None:
                for i in node.idx:
                    if self.point_labels[i] in cond:
                        best.offer(float(self.points[i] @ q), int(i))
                return
            lb = self._upper_bound_mip(q, qn, node.left)
            rb = self._upper_bound_mip(q, qn, node.right)
            first, second = ((node.left, node.right) if lb >= rb
                             else (node.right, node.left))
            dfs(first)
            dfs(second)

        dfs(self.root)
        return [(self.values[i], s) for s, i in best.items()]
