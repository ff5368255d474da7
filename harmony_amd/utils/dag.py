"""Minimal DAG with dependency-driven release (reference: utils/DAGImpl.java:33,
used by the ET plan engine ETPlan.java:36-87 — onComplete returns newly
unblocked vertices)."""

from __future__ import annotations

from typing import Dict, Generic, Hashable, List, Set, TypeVar

T = TypeVar("T", bound=Hashable)


class DAG(Generic[T]):
    def __init__(self):
        self._succ: Dict[T, Set[T]] = {}
        self._pred: Dict[T, Set[T]] = {}

    def add_vertex(self, v: T) -> None:
        self._succ.setdefault(v, set())
        self._pred.setdefault(v, set())

    def add_edge(self, src: T, dst: T) -> None:
        self.add_vertex(src)
        self.add_vertex(dst)
        self._succ[src].add(dst)
        self._pred[dst].add(src)
        if self._has_cycle():
            self._succ[src].discard(dst)
            self._pred[dst].discard(src)
            raise ValueError(f"edge {src}->{dst} creates a cycle")

    def vertices(self) -> List[T]:
        return list(self._succ)

    def roots(self) -> List[T]:
        return [v for v, p in self._pred.items() if not p]

    def on_complete(self, v: T) -> List[T]:
        """Remove v; return vertices that became roots (newly executable)."""
        released = []
        for s in self._succ.pop(v, set()):
            self._pred[s].discard(v)
            if not self._pred[s]:
                released.append(s)
        self._pred.pop(v, None)
        for p in self._pred.values():
            p.discard(v)
        return released

    def empty(self) -> bool:
        return not self._succ

    def _has_cycle(self) -> bool:
        seen: Set[T] = set()
        stack: Set[T] = set()

        def visit(v: T) -> bool:
            if v in stack:
                return True
            if v in seen:
                return False
            seen.add(v)
            stack.add(v)
            bad = any(visit(s) for s in self._succ[v])
            stack.discard(v)
            return bad

        return any(visit(v) for v in list(self._succ))
