"""A short randomized parity slice for the regular GPU suite (the long-form
soak lives in tools/fuzz_parity.py; ~3,800 iterations ran green on hardware
in round 1)."""
import os
import subprocess
import sys

import pytest

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_randomized_parity_slice():
    env = dict(os.environ, FUZZ_SECONDS="15", FUZZ_SEED="1234")
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "fuzz_parity.py")],
        env=env, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "0 failures" in out.stdout
