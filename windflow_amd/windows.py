"""Window / join operator lowering onto the native engine.

Maps the builder descriptors (Keyed/Parallel/Paned/MapReduce/Ffat windows,
Interval_Join — reference wf/keyed_windows.hpp, wf/parallel_windows.hpp,
wf/paned_windows.hpp, wf/mapreduce_windows.hpp, wf/ffat_windows.hpp,
wf/interval_join.hpp) onto the C++ logics in csrc/engine/windows.cpp.

Aggregation spec accepted as the builder `func` (or Ffat comb):
  * ("sum"|"min"|"max"|"count"|"avg", col)  — compiled incremental combine
  * windflow_amd.native.win_agg(op, col)    — same, as a NativeLogic
  * a Python callable(win: dict) -> float   — non-incremental path
    (keyed windows only); win holds 'ts', 'key', 'gwid', 'c0'.. views.

Paned/MapReduce decomposition (reference paned_windows.hpp:83-84,
mapreduce_windows.hpp:83-84) is lowered as two engine operators exchanging
explicit (pane/window id, partial) rows, so correctness does not depend on
channel arrival order; MAP stages use an ORDERING collector to make the
round-robin tuple partitioning consistent across replicas.
"""
import math

from .basic import WinType, JoinMode, CollectorKind, RoutingMode
from .operators import NativeLogic

COMB = {"sum": 0, "min": 1, "max": 2, "count": 3, "avg": 4}


def _parse_agg(fn):
    """-> (comb_id, col, pyfn)."""
    if isinstance(fn, tuple) and len(fn) == 2 and isinstance(fn[0], str):
        return COMB[fn[0]], int(fn[1]), None
    if isinstance(fn, NativeLogic) and fn.kind == "win_agg":
        return COMB[fn.spec], int(fn.iparams[0]), None
    if callable(fn):
        return 0, 0, fn
    raise TypeError(f"window aggregation spec not understood: {fn!r}")


def _use_int(out_schema, comb):
    # integer result path for I64/I32 outputs and non-avg combines
    return 1 if out_schema and out_schema[0] in (0, 4) and comb != COMB["avg"] else 0


def _win_params(op):
    w = op.window
    return int(w["type"]), int(w["win"]), int(w["slide"]), int(w.get("lateness", 0))


def _attach_tail(graph, e, node, out_id):
    # stages chained after the window op (e.g. chain_sink) attach to out_id
    for extra_op in node.ops[1:]:
        a = graph._stage_args(extra_op, node)
        e.chain_stage(out_id, a[0], a[1], fparams=a[2], iparams=a[3],
                      out_schema=a[4], out_batch=a[5], pyfn=a[6])
        if extra_op.kind == "sink":
            graph._sink_map[id(extra_op)] = out_id


def lower_window_node(graph, e, node):
    """Create engine op(s) for one window/join node; returns (in_id, out_id)."""
    op = node.ops[0]
    kind = op.kind
    out_schema = op.out_schema or [0]

    if kind == "interval_join":
        j = op.join
        mode = int(j.get("mode", JoinMode.KP))
        ca = int(j.get("colA", 0))
        cb = int(j.get("colB", ca))
        ip = [mode, int(j["lower"]), int(j["upper"]), ca, cb]
        # user predicate/result callable (vectorized over matched pairs)
        pyfn = op.logic if callable(op.logic) else None
        eid = e.add_op(op.name or kind, node.parallelism, "interval_join",
                       iparams=ip, out_schema=out_schema, out_batch=op.out_batch,
                       pyfn=pyfn)
        if mode == JoinMode.DP:
            node.in_collector = CollectorKind.ORDERING
        in_id = out_id = eid

    elif kind in ("keyed_windows", "parallel_windows", "ffat_windows"):
        wt, win, slide, lat = _win_params(op)
        agg_src = op.logic if op.logic is not None else op.window.get("comb")
        if kind == "ffat_windows" and agg_src is None:
            agg_src = (op.window.get("comb"), op.window.get("lift", 0))
        comb, col, pyfn = _parse_agg(agg_src)
        ui = _use_int(out_schema, comb)
        ekind = {"keyed_windows": "win_keyed", "parallel_windows": "win_parallel",
                 "ffat_windows": "win_ffat"}[kind]
        if pyfn:  # python fns execute on the keyed python engine (all forms)
            ekind = "win_keyed"
        own = 1 if (kind == "parallel_windows" and not pyfn) else 0
        ip = [wt, win, slide, lat, comb, col, own, ui]
        eid = e.add_op(op.name or kind, node.parallelism, ekind, iparams=ip,
                       out_schema=out_schema, out_batch=op.out_batch, pyfn=pyfn)
        in_id = out_id = eid

    elif kind == "paned_windows":
        wt, win, slide, lat = _win_params(op)
        comb, col, pyfn = _parse_agg(op.logic)
        if pyfn:
            # python pane functions can't be recombined by the native WLQ;
            # run the same windows on the keyed python engine instead (same
            # results, key-partitioned rather than pane-partitioned)
            eid = e.add_op(op.name or kind, node.parallelism, "win_keyed",
                           iparams=[wt, win, slide, lat, 0, col, 0, 0],
                           out_schema=out_schema, out_batch=op.out_batch,
                           pyfn=pyfn)
            in_id = out_id = eid
            _attach_tail(graph, e, node, out_id)
            return in_id, out_id
        pane = math.gcd(win, slide)
        win_p, slide_p = win // pane, slide // pane
        ui = _use_int(out_schema, comb)
        plq_par = int(op.extra.get("plq_par", node.parallelism))
        wlq_par = int(op.extra.get("wlq_par", node.parallelism))
        mid_schema = [0, out_schema[0], 0]  # (pane gwid, partial, slice count)
        # Keys are partitioned (KEYBY) into both stages: each replica owns
        # every pane/window of its keys, so a tuple is processed once, not
        # once per replica.  The reference's broadcast + gwid%n ownership
        # (parallelism beyond the key count) remains as Parallel_Windows.
        plq = e.add_op((op.name or kind) + ".plq", plq_par, "win_plq",
                       iparams=[wt, pane, pane, lat, comb, col, 0, ui],
                       out_schema=mid_schema, out_batch=op.out_batch)
        wlq = e.add_op((op.name or kind) + ".wlq", wlq_par, "win_wlq",
                       iparams=[wt, win_p, slide_p, lat, comb, pane, 0, ui],
                       out_schema=out_schema, out_batch=op.out_batch)
        e.add_edge(plq, wlq, RoutingMode.KEYBY, CollectorKind.WATERMARK, -1)
        in_id, out_id = plq, wlq

    elif kind == "mapreduce_windows":
        wt, win, slide, lat = _win_params(op)
        comb, col, pyfn = _parse_agg(op.logic)
        if pyfn:
            eid = e.add_op(op.name or kind, node.parallelism, "win_keyed",
                           iparams=[wt, win, slide, lat, 0, col, 0, 0],
                           out_schema=out_schema, out_batch=op.out_batch,
                           pyfn=pyfn)
            in_id = out_id = eid
            _attach_tail(graph, e, node, out_id)
            return in_id, out_id
        ui = _use_int(out_schema, comb)
        map_par = int(op.extra.get("map_par", node.parallelism))
        red_par = int(op.extra.get("reduce_par", node.parallelism))
        mid_schema = [0, out_schema[0], 0]  # (window gwid, partial, slice count)
        mp = e.add_op((op.name or kind) + ".map", map_par, "win_mr_map",
                      iparams=[wt, win, slide, lat, comb, col, 0, ui],
                      out_schema=mid_schema, out_batch=op.out_batch)
        rd = e.add_op((op.name or kind) + ".reduce", red_par, "win_mr_reduce",
                      iparams=[map_par, comb, ui],
                      out_schema=out_schema, out_batch=op.out_batch)
        e.add_edge(mp, rd, RoutingMode.KEYBY, CollectorKind.WATERMARK, -1)
        # consistent arrival order across MAP replicas (round-robin split)
        node.in_collector = CollectorKind.ORDERING
        in_id, out_id = mp, rd

    else:
        raise NotImplementedError(f"window kind {kind}")

    _attach_tail(graph, e, node, out_id)
    return in_id, out_id
