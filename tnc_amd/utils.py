"""Small data structures, mirroring tnc/src/utils/datastructures.rs."""

from __future__ import annotations


class UnionFind:
    """datastructures.rs:9: path-compressing union-find."""

    def __init__(self, n: int):
        self.parent = list(range(n))
        self.sets = n

    def find(self, x: int) -> int:
        root = x
        while self.parent[root] != root:
            root = self.parent[root]
        while self.parent[x] != root:
            self.parent[x], x = root, self.parent[x]
        return root

    def union(self, a: int, b: int):
        ra, rb = self.find(a), self.find(b)
        if ra != rb:
            self.parent[rb] = ra
            self.sets -= 1

    def count_sets(self) -> int:
        return self.sets
