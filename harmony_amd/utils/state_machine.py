"""Typed state machine (reference: utils/StateMachine.java:27 — used by
WorkerStateManager's INIT->RUN<->OPTIMIZE->RUN_FINISHING->CLEANUP and the
ownership-sync protocol)."""

from __future__ import annotations

import threading
from typing import Dict, Set, Tuple


class StateMachine:
    def __init__(self, states: Set[str], initial: str,
                 transitions: Set[Tuple[str, str]]):
        assert initial in states
        for a, b in transitions:
            assert a in states and b in states
        self._states = states
        self._trans: Dict[str, Set[str]] = {}
        for a, b in transitions:
            self._trans.setdefault(a, set()).add(b)
        self._cur = initial
        self._cv = threading.Condition()

    @property
    def state(self) -> str:
        return self._cur

    def set(self, new: str) -> None:
        with self._cv:
            if new not in self._trans.get(self._cur, set()):
                raise ValueError(f"illegal transition {self._cur} -> {new}")
            self._cur = new
            self._cv.notify_all()

    def compare_and_set(self, expect: str, new: str) -> bool:
        with self._cv:
            if self._cur != expect:
                return False
            if new not in self._trans.get(self._cur, set()):
                raise ValueError(f"illegal transition {self._cur} -> {new}")
            self._cur = new
            self._cv.notify_all()
            return True

    def wait_for(self, state: str, timeout: float = None) -> bool:
        with self._cv:
            return self._cv.wait_for(lambda: self._cur == state,
                                     timeout=timeout)
