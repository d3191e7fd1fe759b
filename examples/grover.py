"""Grover search for a marked item (parity: /root/reference/examples/grovers.cpp)."""
import sys, math
import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import qrack_amd as qa


def grover(n=8, target=0x5A, shots=100):
    target &= (1 << n) - 1
    q = qa.create_simulator(n, seed=42)
    q.set_permutation(0)
    for i in range(n):
        q.h(i)
    iters = int(math.pi / 4 * math.sqrt(2**n))
    for _ in range(iters):
        # oracle: phase flip the target (X-conjugate set bits so |target>
        # maps onto |0...0>, whose phase ZeroPhaseFlip flips)
        for i in range(n):
            if (target >> i) & 1:
                q.x(i)
        q.zero_phase_flip(0, n)
        for i in range(n):
            if (target >> i) & 1:
                q.x(i)
        # diffusion
        for i in range(n):
            q.h(i)
        q.zero_phase_flip(0, n)
        for i in range(n):
            q.h(i)
        q.phase_flip()
    res = q.multi_shot_measure_mask([1 << i for i in range(n)], shots)
    return max(res, key=res.get), res


if __name__ == "__main__":
    best, res = grover()
    print(f"marked item found: 0x{best:X} ({res[best]}% of shots)")
    assert best == 0x5A
