"""Hierarchical tic/toc profiler.

Mirrors the reference's amgcl::profiler (amgcl/profiler.hpp:54): a named
tree of scoped timers printed as a percentage profile.
"""
import time
from contextlib import contextmanager


class _Node:
    __slots__ = ("children", "total", "started")

    def __init__(self):
        self.children = {}
        self.total = 0.0
        self.started = None


class Profiler:
    def __init__(self, name="profile"):
        self.name = name
        self.root = _Node()
        self.stack = [self.root]

    def tic(self, name):
        node = self.stack[-1].children.setdefault(name, _Node())
        node.started = time.perf_counter()
        self.stack.append(node)

    def toc(self, name=None):
        node = self.stack.pop()
        node.total += time.perf_counter() - node.started
        node.started = None

    @contextmanager
    def scope(self, name):
        self.tic(name)
        try:
            yield
        finally:
            self.toc(name)

    def report(self):
        lines = [f"[{self.name}]"]
        grand = sum(c.total for c in self.root.children.values())

        def walk(node, depth):
            for name, child in node.children.items():
                pct = 100.0 * child.total / grand if grand else 0.0
                lines.append(f"{'  ' * depth}{name:<30s} {child.total:10.3f} s ({pct:5.1f}%)")
                walk(child, depth + 1)

        walk(self.root, 1)
        return "\n".join(lines)


prof = Profiler("amgcl_amd")
