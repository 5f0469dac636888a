"""Fluent builder API — parity with the reference's wf/builders.hpp (1691
LoC) and the accepted-signature catalogue in the reference `API` file.

Python adaptation of the signatures: user logic is either
  * a `windflow_amd.native.*` spec (compiled fast path), or
  * a per-batch Python callable over a dict of numpy column views
    ({'ts','key','c0','c1',...,'watermark'}); map mutates in place,
    transform/flatmap returns a new dict, filter returns a bool mask,
    source returns a dict per step (None = end of stream).

`withKeyBy` takes either a payload column index (compiled key extraction)
or a callable(cols)->uint64 array, mirroring the reference's key_extractor
(builders.hpp:217-234 re-typing builders).
"""
from .basic import WinType, JoinMode
from .operators import Operator, NativeLogic


class _BasicBuilder:
    """reference: wf/builders.hpp:57-130 (Basic_Builder)."""
    _kind = None
    _needs_key = False

    def __init__(self, func=None):
        self._op = Operator(kind=self._kind, logic=func,
                            name=f"{self._kind}")

    def withName(self, name):
        self._op.name = name
        return self

    def withParallelism(self, p):
        if int(p) < 1:
            raise ValueError(f"parallelism must be >= 1 (got {p})")
        self._op.parallelism = int(p)
        return self

    def withOutputBatchSize(self, b):
        self._op.out_batch = int(b)
        return self

    def withOutputSchema(self, dtypes):
        """Payload column dtypes of this operator's output (DType ints)."""
        self._op.out_schema = list(dtypes)
        return self

    def withClosingFunction(self, fn):
        self._op.closing = fn
        return self

    def withKeyBy(self, key_extractor):
        if isinstance(key_extractor, int):
            self._op.key_extractor = ('col', key_extractor)
        else:
            self._op.key_extractor = key_extractor
        return self

    def withRebalancing(self):
        self._op.rebalancing = True
        return self

    def withBroadcast(self):
        """BROADCAST routing of inputs (reference builders.hpp:252): every
        replica of this operator receives every input batch."""
        self._op.broadcast_input = True
        return self

    def build(self):
        op = self._op.clone()
        if self._needs_key and op.key_extractor is None:
            # keyed input already carries a key column upstream
            op.key_extractor = 'carried'
        return op


class Source_Builder(_BasicBuilder):
    _kind = "source"


class Map_Builder(_BasicBuilder):
    _kind = "map"


class Filter_Builder(_BasicBuilder):
    _kind = "filter"


class FlatMap_Builder(_BasicBuilder):
    _kind = "flatmap"


class Reduce_Builder(_BasicBuilder):
    """Keyed running aggregate — KEYBY routing in (reference reduce.hpp:285).

    Accepts a native spec (native.keyed_sum_reduce / keyed_reduce_f) or an
    ARBITRARY user function fn(acc, value) -> new_acc applied per tuple in
    key order, per-key state in the replica (the reference's
    void(const tuple_t&, result_t&) signature): emits the updated
    (key, acc) per input tuple."""
    _kind = "reduce"
    _needs_key = True

    def withInitialState(self, v):
        """Initial per-key accumulator value (reference builders.hpp:627)."""
        lg = self._op.logic
        if callable(lg) and not isinstance(lg, NativeLogic):
            self._op.extra['reduce_init'] = v
            return self
        if not (isinstance(lg, NativeLogic) and lg.kind == "reduce"):
            raise TypeError("withInitialState applies to keyed reduces")
        ip = list(lg.iparams)
        while len(ip) < 2:
            ip.append(0)
        ip[1] = int(v)
        self._op.logic = NativeLogic(lg.kind, lg.spec, list(lg.fparams), ip)
        return self

    def build(self):
        op = super().build()
        fn = op.logic
        if callable(fn) and not isinstance(fn, NativeLogic):
            import numpy as np
            state = {}
            init = op.extra.get('reduce_init', 0)
            out_dt = {0: np.int64, 1: np.float64, 2: np.float32,
                      3: np.uint64, 4: np.int32}.get(
                          (op.out_schema or [0])[0], np.float64)

            def reduce_batch(cols, _fn=fn):
                keys = cols['key']
                v = cols['c0']
                out = np.empty(len(keys), dtype=out_dt)
                for i in range(len(keys)):
                    k = int(keys[i])
                    a = _fn(state.get(k, init), v[i])
                    state[k] = a
                    out[i] = a
                return {"ts": cols['ts'], "key": keys, "c0": out}

            op.kind = "flatmap"
            op.logic = reduce_batch
        return op


class Sink_Builder(_BasicBuilder):
    _kind = "sink"


class _WindowsBuilder(_BasicBuilder):
    """Shared window config (reference builders.hpp:743-790)."""

    def __init__(self, func=None, lift=None, comb=None):
        super().__init__(func)
        self._op.window = dict(type=WinType.CB, win=0, slide=0, lateness=0,
                               lift=lift, comb=comb)

    @staticmethod
    def _check_extent(win, slide):
        if win < 1 or slide < 1:
            raise ValueError(
                f"window length and slide must be >= 1 (got win={win}, "
                f"slide={slide})")

    def withCBWindows(self, win_len, slide_len):
        self._check_extent(int(win_len), int(slide_len))
        self._op.window.update(type=WinType.CB, win=int(win_len), slide=int(slide_len))
        return self

    def withTBWindows(self, win_us, slide_us):
        self._check_extent(int(win_us), int(slide_us))
        self._op.window.update(type=WinType.TB, win=int(win_us), slide=int(slide_us))
        return self

    def withLateness(self, lateness_us):
        self._op.window['lateness'] = int(lateness_us)
        return self


class Keyed_Windows_Builder(_WindowsBuilder):
    _kind = "keyed_windows"
    _needs_key = True


class Parallel_Windows_Builder(_WindowsBuilder):
    """BROADCAST input; replica i owns windows w ≡ i (mod n)
    (reference parallel_windows.hpp:194)."""
    _kind = "parallel_windows"

    def __init__(self, func=None, lift=None, comb=None):
        super().__init__(func, lift, comb)
        # python (non-incremental) functions run on the keyed python engine
        # (KEYBY); compiled combines use BROADCAST + gwid%n window ownership
        self._op.broadcast_input = not callable(func)
        if callable(func):
            import warnings
            warnings.warn(
                "Parallel_Windows with a Python callable runs on the keyed "
                "python engine (per-key KEYBY partitioning), not the "
                "reference's whole-stream BROADCAST decomposition; same "
                "windows per key, different replica ownership",
                stacklevel=2)
            self._op.key_extractor = 'carried'


class Paned_Windows_Builder(_WindowsBuilder):
    """PLQ/WLQ pane decomposition, pane = gcd(win, slide)
    (reference paned_windows.hpp:83-84)."""
    _kind = "paned_windows"

    def __init__(self, plq_func=None, wlq_func=None, lift=None, comb=None):
        super().__init__(plq_func, lift, comb)
        self._op.extra['wlq_func'] = wlq_func
        if callable(plq_func):
            import warnings
            warnings.warn(
                "Paned_Windows with a Python callable runs the full windows "
                "on the keyed python engine (no PLQ/WLQ pane decomposition): "
                "same results, key-partitioned instead of pane-partitioned",
                stacklevel=2)
        self._op.key_extractor = 'carried'  # KEYBY into the PLQ stage

    def withPLQParallelism(self, p):
        self._op.extra['plq_par'] = int(p)
        return self

    def withWLQParallelism(self, p):
        self._op.extra['wlq_par'] = int(p)
        return self


class MapReduce_Windows_Builder(_WindowsBuilder):
    _kind = "mapreduce_windows"

    def __init__(self, map_func=None, reduce_func=None, lift=None, comb=None):
        super().__init__(map_func, lift, comb)
        self._op.extra['reduce_func'] = reduce_func
        # python (non-incremental) map functions run on the keyed python
        # engine (KEYBY); compiled combines use the reference's BROADCAST +
        # round-robin MAP decomposition
        self._op.broadcast_input = not callable(map_func)
        if callable(map_func):
            import warnings
            warnings.warn(
                "MapReduce_Windows with a Python callable runs on the keyed "
                "python engine (per-key KEYBY partitioning), not the "
                "reference's round-robin MAP decomposition; same windows per "
                "key, different replica ownership",
                stacklevel=2)
            self._op.key_extractor = 'carried'   # KEYBY on the key column

    def withMAPParallelism(self, p):
        self._op.extra['map_par'] = int(p)
        return self

    def withREDUCEParallelism(self, p):
        self._op.extra['reduce_par'] = int(p)
        return self


class Ffat_Windows_Builder(_WindowsBuilder):
    """FlatFAT aggregator: lift(tuple)->agg, comb(agg,agg)->agg
    (reference ffat_windows.hpp / flatfat.hpp)."""
    _kind = "ffat_windows"
    _needs_key = True

    def __init__(self, lift=None, comb=None):
        super().__init__(None, lift, comb)


class Interval_Join_Builder(_BasicBuilder):
    """reference interval_join.hpp; KP (key-partitioned) or DP (data-parallel)."""
    _kind = "interval_join"
    _needs_key = True

    def __init__(self, func=None):
        super().__init__(func)
        self._op.join = dict(mode=JoinMode.KP, lower=0, upper=0, colA=0, colB=0)

    def withBoundaries(self, lower_us, upper_us):
        self._op.join.update(lower=int(lower_us), upper=int(upper_us))
        return self

    def withKPMode(self):
        self._op.join['mode'] = JoinMode.KP
        self._op.broadcast_input = False
        return self

    def withDPMode(self):
        """Data-parallel join: both streams broadcast, each replica stores
        its round-robin slice (reference interval_join.hpp DP +
        join_collector.hpp)."""
        self._op.join['mode'] = JoinMode.DP
        self._op.broadcast_input = True
        return self

    def withValueCols(self, colA, colB=None):
        """Payload columns joined into the (c0=A, c1=B) output pair."""
        self._op.join['colA'] = int(colA)
        self._op.join['colB'] = int(colA if colB is None else colB)
        return self
