"""Isolated timing of the batched-1q-gate kernel vs sequential gates.
Usage: python tools/batch_probe.py [qubits]
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa

n = int(sys.argv[1]) if len(sys.argv) > 1 else 28
q = qa.create_simulator(n, engine="hip", seed=1)
rng = np.random.default_rng(2)


def u2():
    th, ph, lm = rng.uniform(0, 2 * np.pi, 3)
    c, s = np.cos(th / 2), np.sin(th / 2)
    return [c, -s * np.exp(1j * lm), s * np.exp(1j * ph), c * np.exp(1j * (ph + lm))]


targets = [3, 9, 15, 21]
ms = [u2() for _ in targets]
flat = [complex(x) for m in ms for x in m]
REPS = 20
q.mtrx_1q_batch(targets, flat)
q.finish()
t0 = time.perf_counter()
for _ in range(REPS):
    q.mtrx_1q_batch(targets, flat)
q.finish()
t1 = time.perf_counter()
batch_ms = 1000 * (t1 - t0) / REPS
t0 = time.perf_counter()
for _ in range(REPS):
    for t, m in zip(targets, ms):
        q.mtrx([complex(x) for x in m], t)
q.finish()
t1 = time.perf_counter()
seq_ms = 1000 * (t1 - t0) / REPS
state_gb = (1 << n) * 8 / 1e9
print(f"n={n} batch(4 gates, 1 pass)={batch_ms:.3f} ms "
      f"({2 * state_gb / batch_ms:.2f} TB/s RMW) "
      f"sequential(4 passes)={seq_ms:.3f} ms speedup={seq_ms / batch_ms:.2f}x")
