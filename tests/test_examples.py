"""Run the example programs end-to-end (parity model: the reference's
examples/*.cpp standalone binaries)."""

import os
import subprocess
import sys

import pytest

EXAMPLES = ["grover.py", "teleport.py", "shor.py", "qft_demo.py", "pauli_chain_evolve.py",
            "approx_supremacy.py"]
EXDIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples")


@pytest.mark.parametrize("script", EXAMPLES)
def test_example_runs(script):
    out = subprocess.run(
        [sys.executable, script],
        cwd=EXDIR,
        capture_output=True,
        text=True,
        timeout=300,
        env={**os.environ, "PYTHONPATH": os.path.dirname(EXDIR)},
    )
    assert out.returncode == 0, f"{script}: {out.stdout}\n{out.stderr}"
