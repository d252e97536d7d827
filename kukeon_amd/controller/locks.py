"""Per-cell scope lock manager.

The daemon serves RPCs on a thread-per-connection server while the
reconcile loop runs on its own thread: every cell mutation (start, stop,
kill, delete, recreate, apply-update, reconcile) must hold the cell's
scope lock so a probe→spawn sequence can never interleave with another
verb's probe→spawn on the same cell (double-spawn), and status persists
can never interleave with spec updates.

Reference parity: internal/controller/runner/runner.go:333-340 (per-cell
lock manager keyed by scope). Locks are re-entrant (RLock) because verbs
compose (recreate → delete → kill → start); entries are refcounted and
dropped when idle so the table does not grow with dead cells.
"""
from __future__ import annotations

import contextlib
import functools
import threading
from typing import Dict, Hashable, List, Tuple


class ScopeLocks:
    def __init__(self) -> None:
        self._mu = threading.Lock()
        # key -> [RLock, refcount]
        self._locks: Dict[Hashable, List] = {}

    @contextlib.contextmanager
    def hold(self, key: Hashable):
        with self._mu:
            ent = self._locks.get(key)
            if ent is None:
                ent = [threading.RLock(), 0]
                self._locks[key] = ent
            ent[1] += 1
        ent[0].acquire()
        try:
            yield
        finally:
            ent[0].release()
            with self._mu:
                ent[1] -= 1
                if ent[1] == 0 and self._locks.get(key) is ent:
                    del self._locks[key]

    def held_count(self) -> int:
        with self._mu:
            return len(self._locks)


def cell_scope(*args) -> Tuple[str, str, str, str]:
    """Scope key from either (realm, space, stack, name) or a CellDoc."""
    if len(args) == 1 and hasattr(args[0], "spec"):
        d = args[0]
        return (d.spec.realm_id, d.spec.space_id, d.spec.stack_id,
                d.metadata.name)
    return tuple(args[:4])  # type: ignore[return-value]


def locked_cell(fn):
    """Decorator: hold the cell scope lock for the duration of a
    controller verb whose args are (self, realm, space, stack, name, ...)
    or (self, doc, ...)."""
    @functools.wraps(fn)
    def wrapper(self, *args, **kwargs):
        key = cell_scope(*args)
        with self.cell_locks.hold(key):
            return fn(self, *args, **kwargs)
    return wrapper
