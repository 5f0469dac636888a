"""Window/join operator lowering — native engine support lands in
csrc/engine/windows.cpp; until then this raises with a clear message."""


def lower_window_node(graph, engine, node):
    raise NotImplementedError(
        f"window/join operator '{node.ops[0].kind}' pending native lowering")
