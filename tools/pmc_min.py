import sys
sys.path.insert(0, sys.argv[1])
import qrack_amd as qa
q = qa.create_simulator(26, engine="hip", seed=1)
q.set_permutation(3)
for _ in range(4):
    q.h(20)
q.finish()
print("done")
