"""Operator descriptors produced by the builders (reference: wf/builders.hpp).

A descriptor is a passive spec; PipeGraph flattens descriptors into the
native engine's OpSpec/EdgeSpec lists at run().
"""
from dataclasses import dataclass, field
from typing import Any, Callable, Optional


@dataclass
class NativeLogic:
    kind: str
    spec: str
    fparams: list = field(default_factory=list)
    iparams: list = field(default_factory=list)


@dataclass
class Operator:
    kind: str                       # source/map/filter/flatmap/reduce/sink/window/join/split/...
    name: str = ""
    parallelism: int = 1
    logic: Any = None               # NativeLogic | python callable
    out_schema: list = field(default_factory=lambda: [0])  # DType ints
    out_batch: int = 1024
    key_extractor: Any = None       # None | ('col', idx) | callable(cols)->u64 array
    rebalancing: bool = False
    broadcast_input: bool = False   # Parallel/Paned windows: replicas see all tuples
    closing: Optional[Callable] = None
    device: int = -1                # -1 CPU, >=0 GPU ordinal
    gpu: bool = False
    window: Optional[dict] = None   # dict(win=, slide=, type=CB/TB, lateness=, ...)
    join: Optional[dict] = None     # dict(mode=KP/DP, lower=, upper=)
    n_branches: int = 0             # split operator
    extra: dict = field(default_factory=dict)

    def clone(self):
        import copy
        return copy.copy(self)
