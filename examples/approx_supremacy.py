"""Approximate wide-circuit simulation with honest fidelity accounting.

Runs a Sycamore-style random circuit far past exact single-GPU width by
combining the approximation knobs (parity: the reference's SDRP + ACE
benchmark protocol, test/benchmarks.cpp test_noisy_fidelity_*):

  - ACE     — cap entangled-unit width; elide controls past the cap
  - SDRP    — round near-separable qubits back to product states
  - NCRP    — snap near-Clifford phases into the stabilizer tableau

Every rounding event multiplies into get_unitary_fidelity(), so the
reported fidelity estimate is the price actually paid.
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa


def run(n=40, depth=8, sdrp=0.25, ace=24, seed=7):
    os.environ.setdefault("QRACK_QUNIT_ACE_MAX_QB", str(ace))
    q = qa.create_simulator(n, layers=["qunit", "stabilizer_hybrid", "hybrid"], seed=seed)
    if sdrp > 0:
        q.set_sdrp(sdrp)
    rng = np.random.default_rng(seed)
    sq = ["sqrt_x", "s", "h"]
    t0 = time.perf_counter()
    for layer in range(depth):
        for i in range(n):
            getattr(q, sq[rng.integers(3)])(i)
        for i in range(layer % 2, n - 1, 2):
            th, ph = rng.uniform(0, 2 * np.pi, 2)
            q.fsim(float(th), float(ph), i, i + 1)
    shots = q.multi_shot_measure_mask([1 << i for i in range(min(n, 32))], 16)
    dt = time.perf_counter() - t0
    print(f"n={n} depth={depth} sdrp={sdrp} ace={ace}: {1000*dt:.1f} ms, "
          f"fidelity estimate {q.get_unitary_fidelity():.4g}, "
          f"{len(shots)} distinct samples/16")


if __name__ == "__main__":
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 24
    run(n=n)
