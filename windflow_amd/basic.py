"""Basic enums and defaults (reference: wf/basic.hpp:78-216)."""
from . import _core

ExecutionMode = _core.ExecMode
TimePolicy = _core.TimePolicy
RoutingMode = _core.Routing
CollectorKind = _core.CollectorKind
DType = _core.DType


class WinType:
    CB = 0
    TB = 1


class JoinMode:
    KP = 0
    DP = 1


# defaults (reference: wf/basic.hpp:199-216)
DEFAULT_WM_INTERVAL_USEC = 100_000
DEFAULT_WM_AMOUNT = 1000
DEFAULT_BATCH = 1024
