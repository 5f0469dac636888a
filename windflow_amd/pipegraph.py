"""PipeGraph / MultiPipe — graph composition layer.

Mirrors the reference's wf/pipegraph.hpp (Application Tree, merge/split)
and wf/multipipe.hpp (add/chain/add_sink/select) on top of the native
engine: every MultiPipe operation edits a flat node/edge list; run()
flattens it into _core.Engine OpSpecs (fused chains become engine stages)
and executes one pinned thread per replica.
"""
import json
import time

from . import _core
from .basic import ExecutionMode, TimePolicy, CollectorKind, RoutingMode
from .operators import Operator, NativeLogic


class _Node:
    def __init__(self, op):
        self.ops = [op]              # fused chain of descriptors
        self.parallelism = op.parallelism
        self.device = op.device
        self.n_branches = op.n_branches
        self.engine_id_in = None     # engine op receiving this node's inputs
        self.engine_id_out = None    # engine op producing this node's outputs
        self.in_collector = None     # collector override for incoming edges

    @property
    def last(self):
        return self.ops[-1]


class MultiPipe:
    """A linear pipeline inside the PipeGraph (reference multipipe.hpp:96)."""

    def __init__(self, graph, tails, tags=None):
        self.graph = graph
        self.tails = tails           # list of (node_idx, branch)
        self.tags = tags or [0] * len(tails)   # join stream tag per tail
        self.closed = False

    # ---- internal ----
    def _routing_for(self, op):
        if op.broadcast_input:
            return RoutingMode.BROADCAST
        if op.key_extractor is not None:
            return RoutingMode.KEYBY
        if op.rebalancing:
            return RoutingMode.REBALANCING
        return RoutingMode.FORWARD

    def _attach(self, op, routing=None, tag=-1):
        g = self.graph
        if self.closed:
            raise RuntimeError("MultiPipe already has a sink")
        routing = routing if routing is not None else self._routing_for(op)
        # key extraction runs fused into each upstream node before the shuffle
        # (also for BROADCAST joins/windows — the key column must exist)
        if (routing in (RoutingMode.KEYBY, RoutingMode.BROADCAST)
                and op.key_extractor not in (None, 'carried')):
            for (n, _b) in self.tails:
                g._chain_key_extract(g.nodes[n], op.key_extractor)
        node_idx = g._new_node(op)
        is_join = op.kind == "interval_join"
        for i, (n, b) in enumerate(self.tails):
            t = self.tags[i] if is_join else tag
            g.edges.append(dict(src=n, branch=b, dst=node_idx, routing=routing, tag=t))
        self.tails = [(node_idx, 0)]
        self.tags = [0]
        return node_idx

    # ---- public API (reference multipipe.hpp) ----
    def add(self, op):
        if op.kind == "sink":
            return self.add_sink(op)
        self._attach(op)
        return self

    def chain(self, op):
        """Fuse `op` into the tail replica threads when legal
        (same parallelism + FORWARD routing — reference multipipe.hpp:536)."""
        g = self.graph
        can = (len(self.tails) == 1
               and self._routing_for(op) == RoutingMode.FORWARD
               and g.nodes[self.tails[0][0]].n_branches == 0
               and g.nodes[self.tails[0][0]].parallelism == op.parallelism
               and g.nodes[self.tails[0][0]].device == op.device
               and not g._node_has_consumers(self.tails[0][0]))
        if can:
            g.nodes[self.tails[0][0]].ops.append(op)
        else:
            self._attach(op)
        return self

    def add_sink(self, op):
        self._attach(op)
        self.closed = True
        return self

    def chain_sink(self, op):
        self.chain(op)
        self.closed = True
        return self

    def merge(self, *others):
        """Merge this MultiPipe with others into one (reference
        pipegraph.hpp:308-460 merge shapes).  Tail order tags join
        streams: this pipe's tails are stream A (tag 0), the first merged
        pipe's are stream B (tag 1), ..."""
        def tail_schema(mp):
            n, _ = mp.tails[0]
            return tuple(self.graph.nodes[n].ops[-1].out_schema or [0])

        # the reference enforces identical tuple types at compile time;
        # merging different schemas here would silently reinterpret columns
        base = tail_schema(self)
        for o in others:
            if tail_schema(o) != base:
                raise TypeError(
                    f"merge requires identical output schemas "
                    f"(got {base} vs {tail_schema(o)})")
        tails = list(self.tails)
        tags = [0] * len(self.tails)
        for i, o in enumerate(others):
            tails += o.tails
            tags += [i + 1] * len(o.tails)
            o.closed = True
        return MultiPipe(self.graph, tails, tags)

    def split(self, split_logic, n_branches):
        """Split into n branches by user logic (reference pipegraph.hpp:265).
        split_logic: native.split_mod(col) / native.split_rr(), or a
        callable(cols) returning int32 branch ids (one branch per tuple) or
        a sequence of per-branch bool masks (one tuple to ANY subset of
        branches — reference splitting_emitter.hpp vector<integral_t>)."""
        up = self.graph.nodes[self.tails[0][0]]
        op = Operator(kind="split", logic=split_logic, name="split",
                      parallelism=up.parallelism,
                      n_branches=n_branches,
                      out_schema=up.last.out_schema,
                      out_batch=up.last.out_batch)
        node_idx = self._attach(op)
        self.closed = True
        return self

    def select(self, branch):
        """MultiPipe of one split branch (reference multipipe.hpp select)."""
        n, _ = self.tails[0]
        nb = self.graph.nodes[n].n_branches
        if not (0 <= int(branch) < nb):
            raise IndexError(f"select({branch}): split has {nb} branches")
        return MultiPipe(self.graph, [(n, branch)])

    def split_gpu(self, n_branches, expr=None, ncols=1):
        """Device split.  Without `expr`: whole-batch round-robin (round-1
        behavior).  With `expr`: PER-TUPLE routing by a JIT branch
        expression over (v0.., ts, key) with on-device compaction per
        branch (reference splitting_emitter_gpu.hpp per-branch batches)."""
        if expr is None:
            return self.split(NativeLogic("split", "rr", [], []), n_branches)
        from .native_gpu import gpu_jit_split
        updev = self.graph.nodes[self.tails[0][0]].device  # upstream device
        mp = self.split(gpu_jit_split(expr, ncols), n_branches)
        node = self.graph.nodes[self.tails[0][0]]          # the split node
        node.ops[0].gpu = True
        node.device = updev if updev >= 0 else 0
        node.ops[0].device = node.device
        return mp


class PipeGraph:
    """reference pipegraph.hpp:74."""

    def __init__(self, name="app", mode=ExecutionMode.DEFAULT,
                 time_policy=TimePolicy.EVENT_TIME, tracing=None):
        self.name = name
        self.mode = mode
        self.time_policy = time_policy
        # tracing: None -> honor WF_TRACING env (runtime knob replacing the
        # reference's compile-time -DWF_TRACING_ENABLED)
        import os as _os
        self.tracing = (bool(_os.environ.get("WF_TRACING"))
                        if tracing is None else bool(tracing))
        self._monitor = None
        self.nodes = []
        self.edges = []
        self.engine = None
        self._sink_map = {}          # sink descriptor id -> engine op id
        self._t0 = self._t1 = None
        self._dist = None            # (rank, world, rccl_id bytes)

    def set_dist(self, rank, world, rccl_id):
        """Configure the RCCL communicator for gpu_keyby_exchange stages.
        rccl_id: bytes from _core.rccl_unique_id() on rank 0, broadcast via
        torch.distributed (see windflow_amd.dist.init_from_torch)."""
        self._dist = (int(rank), int(world), rccl_id)
        return self

    # ---- construction ----
    def add_source(self, op):
        idx = self._new_node(op)
        return MultiPipe(self, [(idx, 0)])

    def _new_node(self, op):
        self.nodes.append(_Node(op))
        return len(self.nodes) - 1

    def _node_has_consumers(self, n):
        return any(e['src'] == n for e in self.edges)

    def _chain_key_extract(self, node, key_extractor):
        if key_extractor in (None, 'carried'):
            return
        ke = Operator(kind="key_extract", logic=key_extractor,
                      out_schema=node.last.out_schema,
                      out_batch=node.last.out_batch)
        node.ops.append(ke)

    # ---- flatten to native engine ----
    def _stage_args(self, op, node):
        """(kind, spec, fparams, iparams, schema, out_batch, pyfn) for one stage."""
        import functools
        kind, spec, fp, ip, pyfn = op.kind, "", [], [], None
        if op.kind == "key_extract":
            if isinstance(op.logic, tuple) and op.logic[0] == 'col':
                return ("keyby_col", "", [], [op.logic[1]], op.out_schema, op.out_batch, None)
            fn = op.logic

            def key_map(cols, _fn=fn):
                cols['key'][:] = _fn(cols)
            return ("map", "", [], [], op.out_schema, op.out_batch, key_map)
        if isinstance(op.logic, NativeLogic):
            nl = op.logic
            kind2 = nl.kind if nl.kind else kind
            return (kind2, nl.spec, list(nl.fparams), list(nl.iparams),
                    op.out_schema, op.out_batch, None)
        if callable(op.logic):
            k = {"flatmap": "flatmap", "map": "map", "filter": "filter",
                 "source": "source", "sink": "sink", "transform": "transform",
                 "split": "split"}.get(op.kind)
            if k is None:
                raise NotImplementedError(f"python logic for kind {op.kind}")
            return (k, "", [], [], op.out_schema, op.out_batch, op.logic)
        raise NotImplementedError(f"cannot lower operator kind={op.kind} logic={op.logic!r}")

    def _lower_node(self, e, node):
        """Create the engine op (+ fused stages) for one graph node."""
        from . import windows as _win
        first = node.ops[0]
        if first.kind in ("keyed_windows", "parallel_windows", "ffat_windows",
                          "paned_windows", "mapreduce_windows", "interval_join"):
            return _win.lower_window_node(self, e, node)
        if first.gpu:
            from . import gpu_exec
            return gpu_exec.lower_gpu_node(self, e, node)
        args = self._stage_args(first, node)
        eid = e.add_op(first.name or first.kind, node.parallelism, args[0], args[1],
                       fparams=args[2], iparams=args[3], out_schema=args[4],
                       out_batch=args[5], pyfn=args[6], device=node.device)
        if first.kind == "sink":
            self._sink_map[id(first)] = eid
        for op in node.ops[1:]:
            a = self._stage_args(op, node)
            e.chain_stage(eid, a[0], a[1], fparams=a[2], iparams=a[3],
                          out_schema=a[4], out_batch=a[5], pyfn=a[6])
            if op.kind == "sink":
                self._sink_map[id(op)] = eid
        return eid

    def _check_terminated(self):
        # every leaf must end in a sink (reference PipeGraph::run errors on
        # un-terminated MultiPipes); otherwise results silently vanish
        has_out = set(d["src"] for d in self.edges)
        for i, node in enumerate(self.nodes):
            if i in has_out or node.n_branches:
                continue
            if not any("sink" in op.kind for op in node.ops):
                name = node.ops[-1].name or node.ops[-1].kind
                raise RuntimeError(
                    f"operator '{name}' has no consumers and is not a sink — "
                    f"terminate the MultiPipe with add_sink()/chain_sink()")

    def build_engine(self):
        import os as _os
        self._check_terminated()
        e = _core.Engine()
        e.mode = self.mode
        e.time_policy = self.time_policy
        # runtime knobs replacing the reference's compile-time macros
        # (README.md:32-41: FF_BOUNDED_BUFFER capacity, NO_DEFAULT_MAPPING)
        if _os.environ.get("WFA_QUEUE_CAP"):
            e.queue_capacity = int(_os.environ["WFA_QUEUE_CAP"])
        if _os.environ.get("WFA_PIN_THREADS"):
            e.pin_threads = True
        if self._dist is not None:
            e.set_dist(*self._dist)
        default_ck = {ExecutionMode.DEFAULT: CollectorKind.WATERMARK,
                      ExecutionMode.DETERMINISTIC: CollectorKind.ORDERING,
                      ExecutionMode.PROBABILISTIC: CollectorKind.KSLACK}[self.mode]
        for node in self.nodes:
            ids = self._lower_node(e, node)
            if isinstance(ids, tuple):
                node.engine_id_in, node.engine_id_out = ids
            else:
                node.engine_id_in = node.engine_id_out = ids
        # engine emitter order must follow branch order per source node
        for edge in sorted(self.edges, key=lambda d: (d['src'], d['branch'])):
            dst = self.nodes[edge['dst']]
            ck = dst.in_collector if dst.in_collector is not None else default_ck
            e.add_edge(self.nodes[edge['src']].engine_id_out,
                       dst.engine_id_in,
                       edge['routing'], ck, edge['tag'])
        self.engine = e
        return e

    # ---- execution (reference pipegraph.hpp:610-739) ----
    def run(self):
        e = self.build_engine()
        if self.tracing:
            from .monitoring import MonitoringThread
            self._monitor = MonitoringThread(self).start()
        self._t0 = time.time()
        e.run()
        self._t1 = time.time()
        if self._monitor:
            self._monitor.stop()
            import os as _os
            logdir = _os.environ.get("WF_LOG_DIR")
            if logdir:
                from .monitoring import dump_stats
                dump_stats(self, _os.path.join(logdir, f"{self.name}.json"))
        for node in self.nodes:
            for op in node.ops:
                if op.closing:
                    op.closing()
        return self

    def start(self):
        e = self.build_engine()
        self._t0 = time.time()
        e.start()
        return self

    def prepare(self):
        """Build + spawn + warm replica threads, blocked at a start gate —
        lets benchmarks exclude one-time init from the timed region."""
        e = self.build_engine()
        e.start_gated()
        return self

    def run_gated(self):
        self._t0 = time.time()
        self.engine.open_gate_and_wait()
        self._t1 = time.time()
        return self

    def wait_end(self):
        self.engine.wait()
        self._t1 = time.time()
        return self

    # ---- results / stats ----
    def elapsed(self):
        return (self._t1 or time.time()) - (self._t0 or time.time())

    def sink_sum(self, sink_op):
        return self.engine.sink_sum(self._sink_map[id(sink_op)])

    def sink_sum_f(self, sink_op):
        """f64 accumulator of a float sum sink (native.sum_sink_f)."""
        return self.engine.sink_sum_f(self._sink_map[id(sink_op)])

    def sink_count(self, sink_op):
        return self.engine.sink_count(self._sink_map[id(sink_op)])

    def getNumDroppedTuples(self):
        return self.engine.dropped()

    def getNumThreads(self):
        return sum(n.parallelism for n in self.nodes)

    def stats(self):
        return self.engine.stats() if self.engine else []

    def stats_json(self):
        """Aggregated per-operator stats (reference pipegraph.hpp:464-522)."""
        per_op = {}
        for r in self.stats():
            d = per_op.setdefault(r['name'] + f"#{r['op']}", dict(
                name=r['name'], replicas=0, inputs=0, tuples_in=0,
                outputs=0, tuples_out=0, svc_us=[], kernels=0,
                bytes_h2d=0, bytes_d2h=0))
            d['replicas'] += 1
            d['inputs'] += r['inputs']
            d['tuples_in'] += r['tuples_in']
            d['outputs'] += r['outputs']
            d['tuples_out'] += r['tuples_out']
            d['svc_us'].append(r['svc_us_ewma'])
            d['kernels'] += r['kernels']
            d['bytes_h2d'] += r['bytes_h2d']
            d['bytes_d2h'] += r['bytes_d2h']
        return json.dumps(dict(graph=self.name,
                               mode=str(self.mode), elapsed_s=self.elapsed(),
                               dropped=self.getNumDroppedTuples() if self.engine else 0,
                               operators=list(per_op.values())), indent=1)
