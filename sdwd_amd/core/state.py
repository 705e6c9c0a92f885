"""Worker state machine (ref worker.py:36-41, 719-758).

Same five states as the reference with a legal-transition guard. A rank that
fails mid-job goes UNAVAILABLE (its shard is requeued by the scheduler — an
improvement over the reference, which silently dropped the shard,
worker.py:498-500); a successful health probe flips it back to IDLE.
"""
from __future__ import annotations

import enum
import threading
from typing import Callable, Dict, List, Set


class State(enum.Enum):
    IDLE = "IDLE"
    WORKING = "WORKING"
    INTERRUPTED = "INTERRUPTED"
    UNAVAILABLE = "UNAVAILABLE"
    DISABLED = "DISABLED"


# Legal transitions (ref worker.py:738-743). Self-transitions are allowed.
_LEGAL: Dict[State, Set[State]] = {
    State.IDLE: {State.WORKING, State.UNAVAILABLE, State.DISABLED, State.INTERRUPTED},
    State.WORKING: {State.IDLE, State.INTERRUPTED, State.UNAVAILABLE},
    State.INTERRUPTED: {State.IDLE, State.UNAVAILABLE, State.DISABLED},
    State.UNAVAILABLE: {State.IDLE, State.DISABLED},
    State.DISABLED: {State.IDLE},
}


class IllegalTransition(RuntimeError):
    pass


class StateMachine:
    """Thread-safe guarded state holder with optional transition hooks."""

    def __init__(self, initial: State = State.IDLE) -> None:
        self._state = initial
        self._lock = threading.Lock()
        self._hooks: List[Callable[[State, State], None]] = []

    @property
    def state(self) -> State:
        return self._state

    def on_transition(self, hook: Callable[[State, State], None]) -> None:
        self._hooks.append(hook)

    def can_transition(self, new: State) -> bool:
        return new == self._state or new in _LEGAL[self._state]

    def set(self, new: State, strict: bool = True) -> bool:
        """Transition to ``new``; raise (strict) or refuse on illegal moves."""
        with self._lock:
            old = self._state
            if new == old:
                return True
            if new not in _LEGAL[old]:
                if strict:
                    raise IllegalTransition(f"{old.value} -> {new.value}")
                return False
            self._state = new
        for hook in self._hooks:
            hook(old, new)
        return True

    def force(self, new: State) -> None:
        with self._lock:
            self._state = new
