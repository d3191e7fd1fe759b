"""QFT period finding demo (parity: /root/reference/examples/qft.cpp)."""
import sys
import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa

if __name__ == "__main__":
    n = 10
    q = qa.create_simulator(n, seed=3)
    q.set_permutation(5)
    q.qft(0, n)
    q.iqft(0, n)
    assert q.m_all() == 5
    print("QFT round trip OK at", n, "qubits")
