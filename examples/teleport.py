"""Quantum teleportation (parity: /root/reference/examples/teleport.cpp)."""
import sys
import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa


def teleport(theta=0.7):
    q = qa.create_simulator(3, seed=7)
    q.ry(theta, 0)          # state to teleport on qubit 0
    q.h(1); q.cnot(1, 2)    # Bell pair on (1, 2)
    q.cnot(0, 1); q.h(0)    # Bell measurement basis
    m0, m1 = q.m(0), q.m(1)
    if m1:
        q.x(2)
    if m0:
        q.z(2)
    return q.prob(2)


if __name__ == "__main__":
    import math
    p = teleport(0.7)
    expect = math.sin(0.7 / 2) ** 2
    print(f"teleported P(1) = {p:.6f} (expected {expect:.6f})")
    assert abs(p - expect) < 1e-5
