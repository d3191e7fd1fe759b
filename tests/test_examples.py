"""Run the example programs end-to-end (parity model: the reference's
examples/*.cpp standalone binaries)."""

import os
import subprocess
import sys

import pytest

EXAMPLES = ["grover.py", "teleport.py", "shor.py", "qft_demo.py", "pauli_chain_evolve.py",
            "approx_supremacy.py", "graph_state.py", "vqe_h2.py"]
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


def test_dist_qft_example_two_ranks():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(21000 + os.getpid() % 8000),
         os.path.join(EXDIR, "dist_qft.py"), "10"],
        cwd=os.path.dirname(EXDIR),
        capture_output=True,
        text=True,
        timeout=300,
        env={**os.environ, "PYTHONPATH": os.path.dirname(EXDIR)},
    )
    assert out.returncode == 0, f"{out.stdout}\n{out.stderr}"
    assert "QFT across 2 rank(s)" in out.stdout
