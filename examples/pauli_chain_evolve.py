"""Trotterized time evolution of a transverse-field Ising chain
(parity: /root/reference/examples/ - Hamiltonian/TimeEvolve usage)."""
import sys
import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa

if __name__ == "__main__":
    n, steps, dt = 6, 20, 0.05
    q = qa.create_simulator(n, seed=5)
    hx = [{"target": i, "matrix": [0, 1, 1, 0]} for i in range(n)]  # X field
    for _ in range(steps):
        q.time_evolve(hx, dt)
        for i in range(n - 1):  # ZZ coupling via CNOT-RZ-CNOT
            q.cnot(i, i + 1)
            q.rz(2 * dt, i + 1)
            q.cnot(i, i + 1)
    mag = sum(1 - 2 * q.prob(i) for i in range(n)) / n
    print(f"<Z> after evolution: {mag:.4f}")
