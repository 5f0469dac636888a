"""Lower GPU operator descriptors onto the native engine.

A GPU node's fused stages share one replica thread and one HIP stream.
If any consumer of the node is a CPU operator, a gpu_to_host (D2H) stage
is appended so downstream CPU logic sees host batches.
"""
from .operators import NativeLogic


def _gpu_stage_args(op):
    nl = op.logic
    if not isinstance(nl, NativeLogic) or not nl.kind.startswith("gpu_"):
        raise NotImplementedError(
            f"GPU operator needs a native_gpu spec, got {nl!r}")
    return (nl.kind, nl.spec, list(nl.fparams), list(nl.iparams),
            op.out_schema, op.out_batch, None)


def lower_gpu_node(graph, e, node):
    node_idx = graph.nodes.index(node)
    first = node.ops[0]
    a = _gpu_stage_args(first)
    dev = node.device if node.device >= 0 else 0
    eid = e.add_op(first.name or first.kind, node.parallelism, a[0], a[1],
                   fparams=a[2], iparams=a[3], out_schema=a[4], out_batch=a[5],
                   device=dev)
    if first.kind in ("gpu_count_sink", "sink"):
        graph._sink_map[id(first)] = eid
    for op in node.ops[1:]:
        a = _gpu_stage_args(op)
        e.chain_stage(eid, a[0], a[1], fparams=a[2], iparams=a[3],
                      out_schema=a[4], out_batch=a[5])
        if op.kind in ("gpu_count_sink", "sink"):
            graph._sink_map[id(op)] = eid
    # D2H bridge when a consumer is a CPU node
    consumers = [graph.nodes[ed['dst']] for ed in graph.edges
                 if ed['src'] == node_idx]
    if any(c.device < 0 and not c.ops[0].gpu for c in consumers):
        if node.n_branches:
            # a chained D2H stage would lose per-branch routing: branches
            # must stay on the GPU (or add an explicit gpu_to_host per
            # branch after select())
            raise NotImplementedError(
                "per-tuple gpu split branches must feed GPU operators; "
                "add a gpu_to_host() stage on each branch for CPU consumers")
        last = node.ops[-1]
        e.chain_stage(eid, "gpu_to_host", "", out_schema=last.out_schema,
                      out_batch=last.out_batch)
    return eid
