"""GPU operator lowering — maps GPU operator descriptors onto the HIP
operator pipeline in csrc/engine/gpu_ops.cpp (pending)."""


def lower_gpu_node(graph, engine, node):
    raise NotImplementedError("GPU operators pending native lowering")
