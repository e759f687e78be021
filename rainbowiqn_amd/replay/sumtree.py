"""Array-backed sum tree for proportional prioritized sampling.

Functional parity with the reference SegmentTree (rainbowiqn/memory.py,
SURVEY.md section 3.1): power-of-two leaf count, update / find-by-prefix-sum /
total.  This CPU version is numpy-vectorized (batched update and batched find)
and is also the behavioral reference for the on-GPU HIP sum tree
(rainbowiqn_amd/ops/csrc/replay.hip).
"""

from __future__ import annotations

import numpy as np


class SumTree:
    def __init__(self, capacity: int):
        assert capacity > 0
        self.capacity = int(capacity)
        self.size = 1
        while self.size < self.capacity:
            self.size *= 2
        # nodes[1] is the root; leaves occupy nodes[self.size : self.size + capacity]
        self.nodes = np.zeros(2 * self.size, dtype=np.float64)
        self.max_priority = 1.0  # running max of raw (already ^omega) priorities

    # -- updates -----------------------------------------------------------
    def update(self, indices: np.ndarray, priorities: np.ndarray) -> None:
        """Set leaf priorities at `indices` and repair ancestors (batched)."""
        indices = np.asarray(indices, dtype=np.int64)
        priorities = np.asarray(priorities, dtype=np.float64)
        if indices.size == 0:
            return
        assert priorities.shape == indices.shape
        self.max_priority = max(self.max_priority, float(priorities.max(initial=0.0)))
        pos = indices + self.size
        self.nodes[pos] = priorities  # last-write-wins on duplicate indices
        parents = np.unique(pos // 2)
        while parents[0] >= 1:
            self.nodes[parents] = self.nodes[2 * parents] + self.nodes[2 * parents + 1]
            if parents[0] == 1:
                break
            parents = np.unique(parents // 2)

    # -- queries -----------------------------------------------------------
    def total(self) -> float:
        return float(self.nodes[1])

    def get(self, indices: np.ndarray) -> np.ndarray:
        return self.nodes[np.asarray(indices, dtype=np.int64) + self.size]

    def find(self, values: np.ndarray) -> np.ndarray:
        """Batched prefix-sum descent: value in [0, total) -> leaf index."""
        v = np.asarray(values, dtype=np.float64).copy()
        idx = np.ones_like(v, dtype=np.int64)
        while idx[0] < self.size:  # all indices are at the same level
            left = 2 * idx
            left_sum = self.nodes[left]
            go_right = v >= left_sum
            v -= np.where(go_right, left_sum, 0.0)
            idx = left + go_right
        return idx - self.size
