"""Shor's algorithm factoring 15 (parity: /root/reference/examples/
shors_factoring.cpp), using the coherent modular-exponentiation ALU
(POWModNOut) + the fused QFT."""
import sys, math
from fractions import Fraction
import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa


def shor_order(a, N, n_count=8):
    # counting register [0, n_count) + work register of equal width
    q = qa.create_simulator(2 * n_count, seed=11)
    for i in range(n_count):
        q.h(i)
    q.pow_mod_n_out(a, N, 0, n_count, n_count)  # work := a^x mod N
    q.qft(0, n_count)  # this build's QFT emits bit-reversed order
    phase = q.m_reg(0, n_count)
    phase_rev = int(format(phase, f"0{n_count}b")[::-1], 2)
    frac = Fraction(phase_rev, 1 << n_count).limit_denominator(N)
    return frac.denominator


def factor_15():
    N = 15
    for a in (7, 8, 11, 13, 2):
        if math.gcd(a, N) != 1:
            continue
        for _ in range(10):
            r = shor_order(a, N)
            if r and r % 2 == 0 and pow(a, r, N) == 1:
                f1 = math.gcd(pow(a, r // 2) - 1, N)
                f2 = math.gcd(pow(a, r // 2) + 1, N)
                if 1 < f1 < N:
                    return f1, N // f1
                if 1 < f2 < N:
                    return f2, N // f2
    raise RuntimeError("no factor found")


if __name__ == "__main__":
    f1, f2 = factor_15()
    print(f"15 = {f1} x {f2}")
    assert sorted((f1, f2)) == [3, 5]
