"""Parent-array skeleton topology with joint removal / reindexing.

Same capability as the reference's Skeleton (reference
data/human36m/skeleton.py:11-88, itself from VideoPose3D): a parent-index
array plus left/right joint lists; removing joints re-wires each surviving
joint to its nearest surviving ancestor and compacts indices.
"""
from __future__ import annotations

from typing import List, Sequence

import numpy as np


class Skeleton:
    def __init__(self, parents: Sequence[int], joints_left=None, joints_right=None):
        self._parents = np.asarray(parents, dtype=int).copy()
        self._joints_left = list(joints_left) if joints_left is not None else None
        self._joints_right = list(joints_right) if joints_right is not None else None
        self._compute_metadata()

    def num_joints(self) -> int:
        return len(self._parents)

    def parents(self) -> np.ndarray:
        return self._parents

    def has_children(self) -> np.ndarray:
        return self._has_children

    def children(self) -> List[List[int]]:
        return self._children

    def joints_left(self):
        return self._joints_left

    def joints_right(self):
        return self._joints_right

    def remove_joints(self, joints_to_remove: Sequence[int]) -> List[int]:
        """Drop the given joints; every surviving joint's parent becomes its
        nearest surviving ancestor. Returns the kept (old) indices."""
        remove = set(joints_to_remove)
        n = len(self._parents)
        kept = [j for j in range(n) if j not in remove]

        # walk each parent pointer up past removed ancestors
        for i in range(n):
            while self._parents[i] in remove:
                self._parents[i] = self._parents[self._parents[i]]

        # old index -> new compact index
        new_index = {old: new for new, old in enumerate(kept)}
        self._parents = np.array(
            [
                new_index[p] if p >= 0 else -1
                for j, p in enumerate(self._parents)
                if j not in remove
            ],
            dtype=int,
        )

        if self._joints_left is not None:
            self._joints_left = [new_index[j] for j in self._joints_left if j in new_index]
        if self._joints_right is not None:
            self._joints_right = [new_index[j] for j in self._joints_right if j in new_index]

        self._compute_metadata()
        return kept

    def _compute_metadata(self):
        n = len(self._parents)
        self._has_children = np.zeros(n, dtype=bool)
        self._children: List[List[int]] = [[] for _ in range(n)]
        for i, p in enumerate(self._parents):
            if p != -1:
                self._has_children[p] = True
                self._children[p].append(i)
