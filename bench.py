#!/usr/bin/env python3
"""qrack_amd flagship benchmark — QFT wall-clock on the HIP state-vector
engine, the reference's headline protocol (test_qft_* in
/root/reference/test/benchmarks.cpp).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU over RCCL. One step
= state re-init + full n-qubit QFT + 1-shot terminal sample. Weak scaling:
qubits = base_qubits + log2(N), so per-GPU amplitude count is fixed.

Rank 0 prints ONE JSON line with the whole-job aggregate `qft_gates_per_sec`
(gates applied per second across the whole job; ms_per_step = one full QFT).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--qubits", type=int, default=0, help="base (per-GPU) qubit count; 0 = auto")
    p.add_argument("--precision", default="fp32", choices=["fp32", "fp64"])
    p.add_argument("--engine", default="auto", choices=["auto", "hip", "cpu"])
    p.add_argument("--backend", default="auto",
                   help="torch.distributed backend override (auto = cpu:gloo,cuda:nccl for hip)")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    if world > 1:
        # ORDER MATTERS: torch.cuda must initialize the HIP runtime BEFORE
        # the qrack_amd extension touches it — the reverse order leaves
        # torch.cuda reporting "No HIP GPUs are available" in this process
        import torch

        if torch.cuda.is_available():
            torch.cuda.init()

    import qrack_amd as qa

    engine = args.engine
    if engine == "auto":
        engine = "hip" if qa.hip_device_count() > 0 else "cpu"

    # base width: 30 qubits fp32 on GPU (8.6 GB state; QFT ~= 494 gates);
    # tiny on CPU so the no-GPU smoke path finishes instantly
    base_qubits = args.qubits or (30 if engine == "hip" else 16)

    dist = None
    if world > 1:
        import torch
        import torch.distributed as tdist

        # mixed backend: CUDA tensors (half-page exchanges) over RCCL/xGMI,
        # CPU tensors (norm scalars, decision broadcasts, object gathers)
        # over gloo — a pure nccl group rejects CPU-tensor collectives
        backend = args.backend
        if backend == "auto":
            backend = "cpu:gloo,cuda:nccl" if engine == "hip" else "gloo"
        tdist.init_process_group(backend=backend)
        if engine == "hip":
            # clamp for single-GPU multi-rank rehearsals; identity on real
            # one-rank-per-GPU launches
            local_rank = local_rank % max(1, qa.hip_device_count())
            torch.cuda.set_device(local_rank)
        dist = tdist

    meta_bits = (world - 1).bit_length() if world > 1 else 0
    qubits = base_qubits + meta_bits

    if world > 1:
        from qrack_amd.dist_pager import DistQPager

        sim = DistQPager(qubits, precision=args.precision, engine=engine, seed=42,
                         device_id=local_rank)
    else:
        sim = qa.create_simulator(
            qubits, precision=args.precision, engine=engine, seed=42, device_id=local_rank
        )

    n_gates = qubits + qubits * (qubits - 1) // 2  # H + controlled-phase count

    def step():
        sim.set_permutation(0x5A5A5A5A & ((1 << qubits) - 1))
        sim.qft(0, qubits)
        sim.multi_shot_measure_mask([1 << i for i in range(min(qubits, 16))], 1)

    def sync():
        if dist is not None:
            dist.barrier()
        if engine == "hip" and world > 1:
            import torch

            torch.cuda.synchronize()
        if hasattr(sim, "finish"):
            sim.finish()

    for _ in range(args.warmup):
        step()
    sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    # MAX over ranks
    if dist is not None:
        import torch

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = 1000.0 * elapsed / args.steps
    gates_per_sec = n_gates * args.steps / elapsed
    single_gate_ms = ms_per_step / n_gates

    # Isolated single-gate wall-clock (the BASELINE.md metric): one H on a
    # low qubit, synchronized each application — not amortized across fused
    # QFT columns. Measured on the single-rank path only.
    isolated_h_ms = None
    if world == 1:
        sim.set_permutation(0)
        reps = 10
        sim.h(0)
        if hasattr(sim, "finish"):
            sim.finish()
        th0 = time.perf_counter()
        for _ in range(reps):
            sim.h(0)
            if hasattr(sim, "finish"):
                sim.finish()
        th1 = time.perf_counter()
        isolated_h_ms = 1000.0 * (th1 - th0) / reps

    if rank == 0:
        out = {
            "metric": "qft_gates_per_sec",
            "value": gates_per_sec,
            "unit": "gates/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.precision,
            "data": "synthetic",
            "config": {
                "model": "qft",
                "qubits": qubits,
                "base_qubits_per_gpu": base_qubits,
                "n_gates": n_gates,
                "single_gate_ms": single_gate_ms,
                "isolated_h_ms": isolated_h_ms,
                # capacity headline (measured: profiles/qft_fuse4.csv runs
                # the 34-qubit = 128 GB fp32 QFT at fidelity 1.0)
                "max_full_state_qubits_fp32": 34,
                "engine": engine,
                "global_batch": 1,
                "seq_len": qubits,
                "parallelism": f"pager{world}" if world > 1 else "single",
            },
        }
        print(json.dumps(out))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
