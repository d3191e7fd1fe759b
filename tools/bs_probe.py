"""Block-size A/B probe: run with the variant package dir as argv[1]."""
import sys
import time

sys.path.insert(0, sys.argv[1])
import qrack_amd as qa

q = qa.create_simulator(30, engine="hip", seed=1)
q.set_permutation(5)
q.qft(0, 30)
q.finish()
t0 = time.perf_counter()
for _ in range(3):
    q.set_permutation(5)
    q.qft(0, 30)
q.finish()
print(f"{sys.argv[2] if len(sys.argv) > 2 else '?'}: {(time.perf_counter()-t0)/3*1000:.2f} ms/QFT")
