"""GPU operator builders — parity with the reference's wf/builders_gpu.hpp
(Filter_GPU :100, Map_GPU :225, Reduce_GPU :350, Ffat_Windows_GPU :466).

GPU user logic is a `windflow_amd.native_gpu.*` spec (compiled gfx950
kernel catalog).  Custom device lambdas land via hiprtc in a later round.
withDevice(d) places the operator on GPU ordinal d (default 0).
"""
from .builders import _BasicBuilder, _WindowsBuilder


class _GpuBuilder(_BasicBuilder):
    def __init__(self, func=None):
        super().__init__(func)
        self._op.gpu = True
        self._op.device = 0
        if func is not None and getattr(func, "kind", "").startswith("gpu_"):
            self._op.kind = func.kind

    def withDevice(self, d):
        self._op.device = int(d)
        return self


class Source_GPU_Builder(_GpuBuilder):
    """MI355X extension: device-resident source (generator kernel)."""
    _kind = "gpu_source"


class Map_GPU_Builder(_GpuBuilder):
    _kind = "gpu_map"


class Filter_GPU_Builder(_GpuBuilder):
    _kind = "gpu_filter"


class Reduce_GPU_Builder(_GpuBuilder):
    _kind = "gpu_reduce"


class Sink_GPU_Builder(_GpuBuilder):
    _kind = "gpu_count_sink"


class KeyBy_Exchange_GPU_Builder(_GpuBuilder):
    """MI355X extension: inter-GPU keyby shuffle (RCCL all-to-allv over
    xGMI).  The reference has no multi-GPU shuffle at all; this is the
    MI355X-native replacement for KeyBy_Emitter_GPU's host re-batching
    (reference keyby_emitter_gpu.hpp:594-638) across GPUs."""
    _kind = "gpu_exchange"


class Ffat_Windows_GPU_Builder(_GpuBuilder):
    """reference builders_gpu.hpp:466 (+withNumWinPerBatch :576)."""
    _kind = "gpu_ffat"

    def __init__(self, func=None):
        super().__init__(func)
        self._op.window = dict(type=0, win=0, slide=0, lateness=0)

    def withCBWindows(self, win_len, slide_len):
        self._op.window.update(type=0, win=int(win_len), slide=int(slide_len))
        if self._op.logic is not None:
            ip = self._op.logic.iparams
            ip[2], ip[3] = int(win_len), int(slide_len)
        return self

    def withNumWinPerBatch(self, n):
        self._op.extra['num_win_per_batch'] = int(n)
        return self
