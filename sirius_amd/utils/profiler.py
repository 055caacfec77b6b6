"""Call-tree timers.

Reference behavior: src/core/profiler.hpp:37-56 (PROFILE macros → rt_graph
timer tree, printed at finalize; JSON export via apps/timers). Here a
context-manager/decorator building the same nested tree; on GPU the timers
synchronize only when `sync=True` is requested (cheap wall timers by
default so they can stay on in production).
"""

from __future__ import annotations

import json
import time
from collections import OrderedDict
from contextlib import contextmanager


class _Node:
    __slots__ = ("count", "total", "children")

    def __init__(self):
        self.count = 0
        self.total = 0.0
        self.children = OrderedDict()


class Profiler:
    def __init__(self, sync: bool = False):
        self.root = _Node()
        self._stack = [self.root]
        self.sync = sync
        self.enabled = True

    def _maybe_sync(self):
        if self.sync:
            import torch

            if torch.cuda.is_available():
                torch.cuda.synchronize()

    @contextmanager
    def __call__(self, name: str):
        if not self.enabled:
            yield
            return
        parent = self._stack[-1]
        node = parent.children.setdefault(name, _Node())
        self._stack.append(node)
        self._maybe_sync()
        t0 = time.time()
        try:
            yield
        finally:
            self._maybe_sync()
            node.total += time.time() - t0
            node.count += 1
            self._stack.pop()

    # -- reporting --------------------------------------------------------

    def _walk(self, node, name, depth, lines):
        if depth >= 0:
            lines.append(f"{'  ' * depth}{name:<40s} {node.count:>6d} "
                         f"{node.total:>10.3f}s")
        for k, ch in node.children.items():
            self._walk(ch, k, depth + 1, lines)

    def report(self) -> str:
        lines = [f"{'timer':<42s} {'count':>6s} {'total':>11s}"]
        self._walk(self.root, "", -1, lines)
        return "\n".join(lines)

    def to_dict(self, node=None) -> dict:
        node = node or self.root
        return {k: {"count": ch.count, "total": ch.total,
                    "sub": self.to_dict(ch)}
                for k, ch in node.children.items()}

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), indent=1)


# global profiler instance (reference: global rt_graph timer, profiler.cpp)
profiler = Profiler()
