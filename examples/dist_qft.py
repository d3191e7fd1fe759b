"""Multi-GPU QFT with the distributed pager (one rank per GPU over RCCL).

Launch (N GPUs of one node):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 examples/dist_qft.py [qubits]

Each rank holds one 2^(n - log2 N)-amplitude page in its GPU's HBM3E. Meta
(page-index) qubits cost ONE half-page RCCL exchange per QFT column (the
lazy qubit map realizes the swap); everything else is fused single-pass
column kernels. On CPU-only machines the same script runs over gloo.
"""
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def main():
    import torch
    import torch.distributed as dist

    if torch.cuda.is_available():
        torch.cuda.init()
    import qrack_amd as qa
    from qrack_amd.dist_pager import DistQPager

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = qa.hip_device_count() > 0
    backend = "cpu:gloo,cuda:nccl" if on_gpu else "gloo"
    dist.init_process_group(backend=backend)
    if on_gpu:
        local_rank = local_rank % max(1, qa.hip_device_count())
        torch.cuda.set_device(local_rank)

    n = int(sys.argv[1]) if len(sys.argv) > 1 else (28 if on_gpu else 12)
    sim = DistQPager(n, engine="hip" if on_gpu else "cpu", seed=42, device_id=local_rank)
    x = 0x5A5A5A5A & ((1 << n) - 1)
    sim.set_permutation(x)
    t0 = time.perf_counter()
    sim.qft(0, n)
    sim.finish()
    dist.barrier()
    dt = 1000 * (time.perf_counter() - t0)
    shots = sim.multi_shot_measure_mask([1 << i for i in range(min(n, 16))], 8)
    if rank == 0:
        print(f"{n}-qubit QFT across {world} rank(s): {dt:.1f} ms, "
              f"{len(shots)} distinct samples/8")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
