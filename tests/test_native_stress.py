"""Runs the native C++ engine stress harness (csrc/tests/engine_stress.cpp).
Built plain here; under ThreadSanitizer via
`python build_ext.py --stress --tsan && ./build/engine_stress_tsan` —
race coverage the reference lacks entirely."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_engine_stress_plain():
    subprocess.check_call([sys.executable, "build_ext.py", "--stress"], cwd=ROOT)
    out = subprocess.run([os.path.join(ROOT, "build", "engine_stress"), "2"],
                         capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "engine_stress OK" in out.stdout
