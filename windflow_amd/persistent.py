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
import itertools
import os

from . import _core
from .operators import Operator, NativeLogic
from .builders import _BasicBuilder

# Deterministic state-file naming: user path + tag + operator name (or a
# construction-order index when unnamed).  id(op) (used in round 1) changed
# every run, so withKeepState() could never find its previous log.
_unnamed_seq = itertools.count()


def _state_path(op, tag):
    base = op.extra['state_path'] or "/tmp/wfa_state"
    name = op.name if op.name else f"op{next(_unnamed_seq)}"
    return f"{base}.{tag}.{name}.log"


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
    StateStore; flushed (and, unless keep, deleted — reference db_handle.hpp
    deleteDb semantics) when the graph closes the operator."""

    def __init__(self, fn, path, cache_cap, filter_mode=False, keep=False):
        self.fn = fn
        self.path = path
        self.keep = keep
        self.store = _core.StateStore(path, cache_cap, fresh=not keep)
        self.filter_mode = filter_mode

    def __call__(self, cols):
        return self.fn(cols, self.store)

    def close(self):
        self.store.flush()
        if not self.keep:
            try:
                os.unlink(self.path)
            except OSError:
                pass


class P_Map_Builder(_PersistBuilder):
    """Python persistent map: fn(cols, store) mutates cols in place using
    store.get/put (reference p_map.hpp get-modify-put)."""
    _kind = "map"

    def build(self):
        op = super().build()
        path = _state_path(op, "pmap")
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'],
                              keep=op.extra['keep'])
        op.logic = wrapped
        op.closing = wrapped.close
        return op


class P_Filter_Builder(_PersistBuilder):
    """Python persistent filter: fn(cols, store) -> bool mask."""
    _kind = "filter"

    def build(self):
        op = super().build()
        path = _state_path(op, "pfil")
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'], True,
                              keep=op.extra['keep'])
        op.logic = wrapped
        op.closing = wrapped.close
        return op


class P_Sink_Builder(_PersistBuilder):
    """Python persistent sink: fn(cols, store)."""
    _kind = "sink"

    def build(self):
        op = super().build()
        path = _state_path(op, "psnk")
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'],
                              keep=op.extra['keep'])
        op.logic = wrapped
        op.closing = wrapped.close
        return op


class P_FlatMap_Builder(_PersistBuilder):
    """Python persistent flatmap: fn(cols, store) -> dict of output columns
    (any length) or None (reference p_flatmap.hpp)."""
    _kind = "flatmap"

    def build(self):
        op = super().build()
        path = _state_path(op, "pfm")
        wrapped = _PyStateful(op.logic, path, op.extra['cache_capacity'],
                              keep=op.extra['keep'])
        op.logic = wrapped
        op.closing = wrapped.close
        return op


class _PKeyedWin:
    """Store-backed keyed windows (reference p_keyed_windows.hpp): each key's
    pending tuples live as ONE serialized record in the persistent store
    ([i64 ts | f64 val] pairs), so window state can exceed host memory and —
    with keep — survive the graph.  CB fires on count, TB on watermark
    passage; incomplete windows flush at EOS (same semantics as the in-memory
    keyed windows, csrc/engine/windows.cpp)."""

    def __init__(self, fn, path, cache_cap, wintype, win, slide, keep=False):
        import numpy as _np
        self.np = _np
        self.fn = fn
        self.path = path
        self.keep = keep
        self.store = _core.StateStore(path, cache_cap, fresh=not keep)
        self.tb = wintype == 1
        self.win, self.slide = int(win), int(slide)
        self.keys = {}              # replica -> keys with pending state
        self.max_ts = {}
        self.next_q = {}            # TB: first not-yet-fired window per key

    def _load(self, k):
        blob = self.store.get(k)
        if not blob:
            return (self.np.empty(0, self.np.int64),
                    self.np.empty(0, self.np.float64))
        a = self.np.frombuffer(blob, self.np.int64)
        n = len(a) // 2
        return a[:n].copy(), a[n:].view(self.np.float64).copy()

    def _save(self, k, ts, vals):
        self.store.put(k, self.np.concatenate(
            [ts, vals.view(self.np.int64)]).tobytes())

    def _fire_cb(self, k, ts, vals, out, at_eos=False):
        W, S = self.win, self.slide
        while len(vals) >= W or (at_eos and len(vals) > 0):
            w = vals[:W]
            out.append((ts[min(len(ts), W) - 1], k, float(self.fn(w))))
            ts, vals = ts[S:], vals[S:]
            if at_eos and len(vals) == 0:
                break
        return ts, vals

    @staticmethod
    def _first_q(t, W, S):
        return max(0, (int(t) - W) // S + 1) if int(t) >= W else 0

    def _fire_tb(self, k, ts, vals, wm, out):
        # fire every absolute-grid window [q*S, q*S+W) with end <= wm whose
        # start is not beyond the key's newest tuple, in order; q is a
        # persistent per-key cursor, so data-free gaps fire as 0 (same as
        # the in-memory engine) and nothing re-fires
        W, S = self.win, self.slide
        mt = self.max_ts.get(k, -1)
        q = self.next_q.get(k)
        if q is None:
            if not len(ts):
                return ts, vals
            q = self._first_q(ts[0], W, S)
        while q * S <= mt:
            end = q * S + W
            if end > wm:
                break
            if len(ts):
                m = (ts >= q * S) & (ts < end)
                out.append((end - 1, k,
                            float(self.fn(vals[m])) if m.any() else 0.0))
            else:
                out.append((end - 1, k, 0.0))
            q += 1
            if len(ts):
                keep = ts >= q * S
                ts, vals = ts[keep], vals[keep]
        self.next_q[k] = q
        return ts, vals

    def __call__(self, cols):
        np = self.np
        ts_in, key_in, v_in = cols['ts'], cols['key'], cols['c0']
        wm = cols['watermark']
        my_keys = self.keys.setdefault(cols.get('replica', 0), set())
        out = []
        order = np.argsort(key_in, kind='stable')
        ks = key_in[order]
        bounds = np.flatnonzero(np.r_[True, ks[1:] != ks[:-1], True])
        for i in range(len(bounds) - 1):
            sel = order[bounds[i]:bounds[i + 1]]
            k = int(ks[bounds[i]])
            ts, vals = self._load(k)
            ts = np.concatenate([ts, ts_in[sel]])
            vals = np.concatenate([vals, v_in[sel].astype(np.float64)])
            if self.tb:
                o = np.argsort(ts, kind='stable')
                ts, vals = ts[o], vals[o]
                # drop tuples entirely before the fired horizon (late in
                # DEFAULT mode, reference window_replica.hpp lateness gate)
                keep = ts >= (self.next_q.get(k) or 0) * self.slide
                ts, vals = ts[keep], vals[keep]
                if not len(ts):
                    self._save(k, ts, vals)
                    my_keys.add(k)
                    continue
                self.max_ts[k] = max(self.max_ts.get(k, 0), int(ts[-1]))
                ts, vals = self._fire_tb(k, ts, vals, wm, out)
            else:
                ts, vals = self._fire_cb(k, ts, vals, out)
            self._save(k, ts, vals)
            my_keys.add(k)
        return self._pack(out)

    def _pack(self, out):
        if not out:
            return None
        np = self.np
        return {"ts": np.array([o[0] for o in out], np.int64),
                "key": np.array([o[1] for o in out], np.uint64),
                "c0": np.array([o[2] for o in out], np.float64)}

    def on_eos(self, replica=0):
        out = []
        for k in sorted(self.keys.get(replica, ())):
            ts, vals = self._load(k)
            if not len(vals):
                continue
            if self.tb:
                # every window containing data has end <= max_ts + win
                ts, vals = self._fire_tb(k, ts, vals,
                                         self.max_ts.get(k, 0) + self.win + 1,
                                         out)
            else:
                ts, vals = self._fire_cb(k, ts, vals, out, at_eos=True)
            self.store.put(k, b"")
        self.store.flush()
        return self._pack(out)

    def close(self):
        self.store.flush()
        if not self.keep:
            try:
                os.unlink(self.path)
            except OSError:
                pass


class P_Keyed_Windows_Builder(_PersistBuilder):
    """Keyed windows whose per-key archives live in the persistent store
    (reference p_keyed_windows.hpp over DBHandle).  fn(values) -> scalar."""
    _kind = "flatmap"

    def __init__(self, func):
        super().__init__(func)
        self._op.window = dict(type=0, win=0, slide=0)

    def withCBWindows(self, win, slide):
        if int(win) < 1 or int(slide) < 1:
            raise ValueError("window length and slide must be >= 1")
        self._op.window = dict(type=0, win=int(win), slide=int(slide))
        return self

    def withTBWindows(self, win_us, slide_us):
        if int(win_us) < 1 or int(slide_us) < 1:
            raise ValueError("window length and slide must be >= 1")
        self._op.window = dict(type=1, win=int(win_us), slide=int(slide_us))
        return self

    def build(self):
        op = super().build()
        w = op.window
        op.window = None  # handled here, not by the graph's window lowering
        path = _state_path(op, "pkw")
        wrapped = _PKeyedWin(op.logic, path, op.extra['cache_capacity'],
                             w['type'], w['win'], w['slide'],
                             keep=op.extra['keep'])
        op.logic = wrapped
        op.closing = wrapped.close
        return op
