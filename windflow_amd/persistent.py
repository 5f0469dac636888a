"""Persistent operators: keyed state out of core.

Parity with the reference's wf/persistent/ tier (P_Filter/P_Map/P_Reduce/
P_Sink over RocksDB DBHandle), re-based on the engine's
embedded VarKV log + LRU write-back cache (csrc/engine/persist.cpp) since
RocksDB is not in the image.  Semantics preserved: per-key get-modify-put
with user serialize/deserialize; the store is a larger-than-memory state
tier, destroyed with the graph unless `keep=True` (reference db_handle.hpp
deleteDb).

Usage:
    P_Reduce_Builder(col=0).withStatePath('/tmp/state').build()   # native
    P_Map_Builder(fn).withStatePath(...)                          # python:
        fn(cols, store) -> None, with store.get(key)/store.put(key, bytes)
"""
import os

from . import _core
from .operators import Operator, NativeLogic
from .builders import _BasicBuilder


class _PersistBuilder(_BasicBuilder):
    _needs_key = True

    def __init__(self, func=None):
        super().__init__(func)
        self._op.extra['state_path'] = None
        self._op.extra['cache_capacity'] = 1 << 16
        self._op.extra['keep'] = False

    def withStatePath(self, path):
        self._op.extra['state_path'] = path
        return self

    def withCacheCapacity(self, n):
        self._op.extra['cache_capacity'] = int(n)
        return self

    def withKeepState(self):
        self._op.extra['keep'] = True
        return self


class P_Reduce_Builder(_PersistBuilder):
    """Persistent keyed running i64 sum (native)."""
    _kind = "p_reduce"

    def __init__(self, col=0):
        super().__init__(None)
        self._op.logic = NativeLogic("p_reduce", "", [], [int(col), 1 << 16])

    def build(self):
        op = super().build()
        path = op.extra['state_path'] or "/tmp/wfa_state"
        op.logic = NativeLogic("p_reduce", path, [],
                               [op.logic.iparams[0], op.extra['cache_capacity'],
                                1 if op.extra['keep'] else 0])
        return op


class _PyStateful:
    """Wraps a user fn(cols, store) into a per-batch map callable holding a
    StateStore; flushed when the graph closes the operator."""

    def __init__(self, fn, path, cache_cap, filter_mode=False):
        self.fn = fn
        self.store = _core.StateStore(path, cache_cap)
        self.filter_mode = filter_mode

    def __call__(self, cols):
        return self.fn(cols, self.store)


class P_Map_Builder(_PersistBuilder):
    """Python persistent map: fn(cols, store) mutates cols in place using
    store.get/put (reference p_map.hpp get-modify-put)."""
    _kind = "map"

    def build(self):
        op = super().build()
        path = (op.extra['state_path'] or "/tmp/wfa_state") + f".pmap.{id(op)}.log"
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'])
        op.logic = wrapped
        op.closing = wrapped.store.flush
        return op


class P_Filter_Builder(_PersistBuilder):
    """Python persistent filter: fn(cols, store) -> bool mask."""
    _kind = "filter"

    def build(self):
        op = super().build()
        path = (op.extra['state_path'] or "/tmp/wfa_state") + f".pfil.{id(op)}.log"
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'], True)
        op.logic = wrapped
        op.closing = wrapped.store.flush
        return op


class P_Sink_Builder(_PersistBuilder):
    """Python persistent sink: fn(cols, store)."""
    _kind = "sink"

    def build(self):
        op = super().build()
        path = (op.extra['state_path'] or "/tmp/wfa_state") + f".psnk.{id(op)}.log"
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'])
        op.logic = wrapped
        op.closing = wrapped.store.flush
        return op
