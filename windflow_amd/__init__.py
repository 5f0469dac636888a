"""windflow_amd — MI355X-native streaming dataflow engine.

A from-scratch re-design of the WindFlow programming model (MultiPipe /
PipeGraph builder API, CPU+GPU operator set, FlatFAT windowed aggregation)
for AMD MI355X: a native C++ runtime (pinned threads + SPSC queues), SoA
micro-batches as the universal message unit, hand-written HIP/CDNA4 kernels
for the GPU operators, and RCCL-over-xGMI for multi-GPU shuffles.
"""
# PyTorch-ROCm bundles its own HIP runtime (libamdhip64.so, SONAME .so.7).
# If our engine loads /opt/rocm's copy first, a later `import torch` loads a
# SECOND runtime into the process and the GPU context corrupts.  Importing
# torch first makes every later libamdhip64.so.7 request (ours included)
# resolve to the one already-loaded runtime.
try:  # pragma: no cover
    import torch  # noqa: F401
except ImportError:
    pass

from .basic import (ExecutionMode, TimePolicy, WinType, JoinMode, RoutingMode, DType)  # noqa: F401

__version__ = "0.1.0"

from .builders import (  # noqa: F401,E402
    Source_Builder, Map_Builder, Filter_Builder, FlatMap_Builder,
    Reduce_Builder, Sink_Builder, Keyed_Windows_Builder,
    Parallel_Windows_Builder, Paned_Windows_Builder,
    MapReduce_Windows_Builder, Ffat_Windows_Builder, Interval_Join_Builder)
from .pipegraph import PipeGraph, MultiPipe  # noqa: F401,E402
from . import native  # noqa: F401,E402
from .persistent import (  # noqa: F401,E402
    P_Reduce_Builder, P_Map_Builder, P_Filter_Builder, P_Sink_Builder,
    P_FlatMap_Builder, P_Keyed_Windows_Builder)
from .kafka import (  # noqa: F401,E402
    Connector_Source_Builder, Kafka_Source_Builder, Kafka_Sink_Builder)
