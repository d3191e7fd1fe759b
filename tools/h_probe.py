"""Isolated gate-kernel throughput probe: times batches of H / CPhase /
ramp on a single HIP engine and prints effective HBM bandwidth."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import qrack_amd as qa

n = int(sys.argv[1]) if len(sys.argv) > 1 else 30
q = qa.create_simulator(n, engine="hip", seed=1)
q.h(0); q.finish()
state_gb = (1 << n) * 8 / 1e9

def timeit(label, fn, reps, bytes_per_op):
    fn()  # warm
    q.finish()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    q.finish()
    dt = (time.perf_counter() - t0) / reps
    print(f"{label}: {dt*1e3:.3f} ms, {bytes_per_op/dt/1e12:.2f} TB/s")

i = [0]
def h_gate():
    q.h(i[0] % n); i[0] += 1
def cphase():
    q.cphase_root_n(3, (i[0]) % n, (i[0] + 7) % n); i[0] += 1
def x_gate():
    q.x(i[0] % n); i[0] += 1
def qft_once():
    q.qft(0, n)

timeit("H (pair kernel, RW full state)", h_gate, 20, 2 * state_gb * 1e9)
timeit("CPhase (one-sided quarter)", cphase, 20, 0.5 * state_gb * 1e9)
timeit("X (xmask swap)", x_gate, 20, 2 * state_gb * 1e9)
timeit("full QFT", qft_once, 3, 0)
