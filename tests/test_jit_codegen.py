"""JIT fold codegen checks (no GPU needed).

The generalized user-fold path (csrc/engine/gpu_jit.cpp) GENERATES hiprtc
kernel source around user C expressions.  These tests validate the
generated source by cross-compiling it for gfx950 with hipcc — the same
front-end hiprtc uses — so codegen/syntax bugs are caught in the CPU tier
before any GPU run.  Numerics are covered by the gpu-marked oracle tests
in test_gpu_ops.py.
"""
import os
import subprocess
import tempfile

import pytest

from windflow_amd import _core, native_gpu

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

pytestmark = pytest.mark.skipif(not os.path.exists(HIPCC),
                                reason="hipcc not available")


def _src_for(logic):
    return _core.debug_jit_fold_source(logic.kind, logic.spec,
                                       list(map(float, logic.fparams)),
                                       list(map(int, logic.iparams)))


def _hipcc_compiles(src):
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "jit.hip")
        with open(p, "w") as f:
            f.write('#include "hip/hip_runtime.h"\n')
            f.write(src)
        r = subprocess.run(
            [HIPCC, "-x", "hip", "--offload-arch=gfx950", "-fsyntax-only",
             "-Wno-unused-variable", p],
            capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, f"generated source rejected:\n{r.stderr}\n--- source ---\n{src}"


def test_avg_reduce_source_compiles():
    _hipcc_compiles(_src_for(native_gpu.gpu_avg_reduce(col=0)))


def test_minmax_window_source_compiles():
    # non-invertible 2-field fold with 2 outputs over 2 value columns
    logic = native_gpu.gpu_jit_ffat_windows(
        1000, 100, lift="fminf(v0, v1);fmaxf(v0, v1)",
        comb="fminf(a0, b0);fmaxf(a1, b1)", finalize="f0;f1",
        identity=(float("inf"), float("-inf")), cols=(0, 1))
    _hipcc_compiles(_src_for(logic))


def test_avg_tb_window_source_compiles():
    logic = native_gpu.gpu_avg_ffat_windows(1000, 100, col=0, tb=True,
                                            lateness=50)
    _hipcc_compiles(_src_for(logic))


def test_stateful_source_compiles():
    m = native_gpu.gpu_jit_stateful_map("s0 = s0 + v0; v0 = s0")
    _hipcc_compiles(_src_for(m))
    f = native_gpu.gpu_jit_stateful_filter("keep = v0 != s0; s0 = v0")
    _hipcc_compiles(_src_for(f))


def test_spec_validation():
    with pytest.raises(ValueError):
        native_gpu.gpu_jit_reduce(lift="v0;1.0f", comb="a0+b0",
                                  identity=(0.0, 0.0))
    with pytest.raises(ValueError):
        native_gpu.gpu_jit_ffat_windows(0, 100)


def test_builder_comb_expr_sets_extent():
    from windflow_amd.builders_gpu import Ffat_Windows_GPU_Builder
    b = (Ffat_Windows_GPU_Builder(comb="fmaxf(a0, b0)",
                                  identity=(float("-inf"),))
         .withCBWindows(500, 50).withOutputSchema([2]))
    op = b.build()
    ip = op.logic.iparams
    assert op.logic.kind == "gpu_jit_ffat"
    assert ip[8] == 500 and ip[9] == 50 and ip[10] == 0
    b2 = (Ffat_Windows_GPU_Builder(comb="a0+b0")
          .withTBWindows(1000, 100).withLateness(30).withOutputSchema([2]))
    op2 = b2.build()
    ip2 = op2.logic.iparams
    assert ip2[8] == 1000 and ip2[9] == 100 and ip2[10] == 1 and ip2[11] == 30


def test_split_source_compiles():
    logic = native_gpu.gpu_jit_split("key % 3", ncols=1)
    _hipcc_compiles(_src_for(logic))
    logic2 = native_gpu.gpu_jit_split("v0 > 0.5f ? 1 : 0", ncols=2)
    _hipcc_compiles(_src_for(logic2))


def test_geomean_window_source_compiles():
    # device intrinsics (__logf/__expf) in user fold expressions
    logic = native_gpu.gpu_jit_ffat_windows(
        1000, 100, lift="__logf(v0);1.0f", comb="a0+b0;a1+b1",
        finalize="(f1 > 0.0f) ? __expf(f0 / f1) : 0.0f",
        identity=(0.0, 0.0), invertible=True)
    _hipcc_compiles(_src_for(logic))


def test_f64_accumulator_sources_compile():
    # double-precision accumulators: reduce + CB + TB variants
    r = native_gpu.gpu_jit_reduce(lift="v0", comb="a0+b0", finalize="f0",
                                  acc="f64")
    _hipcc_compiles(_src_for(r))
    w = native_gpu.gpu_jit_ffat_windows(
        1000, 100, lift="v0;1.0", comb="a0+b0;a1+b1",
        finalize="(f1 > 0.0) ? (f0 / f1) : 0.0", identity=(0.0, 0.0),
        invertible=True, acc="f64")
    _hipcc_compiles(_src_for(w))
    t = native_gpu.gpu_jit_ffat_windows(
        1000, 100, comb="fmin(a0, b0)", identity=(float("inf"),),
        tb=True, lateness=10, acc="f64")
    _hipcc_compiles(_src_for(t))


def test_f64_flag_validation():
    with pytest.raises(ValueError):
        native_gpu.gpu_jit_reduce(acc="f16")
