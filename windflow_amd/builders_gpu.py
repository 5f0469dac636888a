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
    """reference builders_gpu.hpp:466 (+withNumWinPerBatch :576).

    Accepts either a compiled native_gpu.* spec (func) or an arbitrary
    user fold via keyword expressions (reference's arbitrary lift/comb
    __device__ functors):
        Ffat_Windows_GPU_Builder(comb="fminf(a0,b0)", identity=(float('inf'),))
        Ffat_Windows_GPU_Builder(lift="v0;1.0f", comb="a0+b0;a1+b1",
                                 finalize="f0/f1", identity=(0, 0),
                                 invertible=True)          # AVG
    then .withCBWindows/withTBWindows sets the extent."""
    _kind = "gpu_ffat"

    def __init__(self, func=None, lift=None, comb=None, finalize=None,
                 identity=None, cols=(0,), invertible=False,
                 max_keys=1 << 16, acc="f32"):
        if func is None and comb is not None:
            from . import native_gpu
            lift = lift if lift is not None else "v0"
            finalize = finalize if finalize is not None else "f0"
            identity = identity if identity is not None else (0.0,)
            func = native_gpu.gpu_jit_ffat_windows(
                1, 1, lift=lift, comb=comb, finalize=finalize,
                identity=identity, cols=cols, invertible=invertible,
                max_keys=max_keys, acc=acc)
        super().__init__(func)
        self._op.window = dict(type=0, win=0, slide=0, lateness=0)

    # iparam offsets of (win, slide, wintype, lateness) per logic kind
    _IP = {"gpu_ffat": (2, 3, 6, 7), "gpu_jit_ffat": (8, 9, 10, 11)}

    def _set_win(self, wt, win, slide):
        self._op.window.update(type=wt, win=int(win), slide=int(slide))
        if self._op.logic is not None:
            iw, isl, it, _ = self._IP[self._op.logic.kind]
            ip = self._op.logic.iparams
            ip[iw], ip[isl], ip[it] = int(win), int(slide), wt

    def withCBWindows(self, win_len, slide_len):
        self._set_win(0, win_len, slide_len)
        return self

    def withTBWindows(self, win_us, slide_us):
        self._set_win(1, win_us, slide_us)
        return self

    def withLateness(self, lateness_us):
        self._op.window['lateness'] = int(lateness_us)
        if self._op.logic is not None:
            _, _, _, il = self._IP[self._op.logic.kind]
            self._op.logic.iparams[il] = int(lateness_us)
        return self

    def withNumWinPerBatch(self, n):
        self._op.extra['num_win_per_batch'] = int(n)
        return self

    def withDenseKeys(self):
        """User-asserted integer keys in [0, max_keys): slot = key, the
        hash probe is skipped (loud overflow check)."""
        lg = self._op.logic
        if lg is None or lg.kind not in ("gpu_ffat", "gpu_jit_ffat"):
            raise ValueError("withDenseKeys: gpu_ffat / gpu_jit_ffat only")
        if lg.kind == "gpu_ffat":
            lg.iparams[9] = 1
        else:
            while len(lg.iparams) < 15:
                lg.iparams.append(0)
            lg.iparams[14] = 1
        return self
