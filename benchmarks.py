#!/usr/bin/env python3
"""qrack_amd benchmark harness — the reference's benchmarkLoopVariable
protocol (/root/reference/test/benchmarks.cpp:98-290): sweep qubit widths
min..max, N samples per width, wall-clock around workload + terminal
measurement, CSV output (avg/std/min/max ms + fidelity estimate).

Workloads (reference case names in parens):
  qft            (test_qft_permutation_init)
  ghz            (test_ghz)
  random_circuit (test_random_circuit_sampling_nn)
  supremacy      (test_quantum_supremacy, Sycamore-style fsim + sqrt gates)
  qv             (quantum volume: depth == width, random SU(4)-ish layers)

Usage:
  python benchmarks.py --workload qft --min-qubits 4 --max-qubits 24 \
      --samples 10 --layers hip --out profiles/qft_hip.csv
"""

import argparse
import sys
import time

import numpy as np

sys.path.insert(0, ".")
import qrack_amd as qa


def run_qft(q, n, rng, depth):
    q.set_permutation(int(rng.integers(1 << min(n, 62))))
    q.qft(0, n)


def run_qft_cosmology(q, n, rng, depth):
    # the reference's test_qft_cosmology: QFT of a fully superposed register
    q.set_permutation(0)
    mats = [_M1Q["h"]] * n
    _apply_1q_layer(q, range(n), mats)
    q.qft(0, n)


def run_clifford(q, n, rng, depth):
    # wide random Clifford circuit (reference QUnitClifford regime: hundreds
    # of qubits on the tableau); run with --layers qunit,stabilizer
    q.set_permutation(0)
    d = depth or 20
    names = ["h", "s", "x", "z", "sqrt_x"]
    for _ in range(d):
        for i in range(n):
            getattr(q, names[rng.integers(len(names))])(i)
        for i in range(0, n - 1, 2):
            if rng.integers(2):
                q.cnot(i, i + 1)
            else:
                q.cz(i, i + 1)


def run_ghz(q, n, rng, depth):
    q.set_permutation(0)
    q.h(0)
    for i in range(n - 1):
        q.cnot(i, i + 1)


_SQRT2 = 0.7071067811865476
_M1Q = {
    "h": [_SQRT2, _SQRT2, _SQRT2, -_SQRT2],
    "x": [0, 1, 1, 0],
    "t": [1, 0, 0, np.exp(1j * np.pi / 4)],
    "s": [1, 0, 0, 1j],
    "sqrt_x": [0.5 + 0.5j, 0.5 - 0.5j, 0.5 - 0.5j, 0.5 + 0.5j],
}


def _u_mtrx(th, ph, lm):
    c, s = np.cos(th / 2), np.sin(th / 2)
    return [c, -s * np.exp(1j * lm), s * np.exp(1j * ph), c * np.exp(1j * (ph + lm))]


def _apply_1q_layer(q, targets, mats):
    """One layer of independent 1q gates → a single fused Mtrx1qBatch pass on
    state-vector engines (layered sims lower it per-gate)."""
    if hasattr(q, "mtrx_1q_batch"):
        q.mtrx_1q_batch(list(targets), [complex(x) for m in mats for x in m])
    else:
        for t, m in zip(targets, mats):
            q.mtrx([complex(x) for x in m], t)


def _apply_cnot_layer(q, controls, targets):
    if hasattr(q, "cnot_batch"):
        q.cnot_batch(list(controls), list(targets))
    else:
        for c, t in zip(controls, targets):
            q.cnot(c, t)


def run_random_circuit(q, n, rng, depth):
    q.set_permutation(0)
    d = depth or n
    names = ["h", "x", "t"]
    for _ in range(d):
        mats = [_M1Q[names[rng.integers(3)]] for _ in range(n)]
        _apply_1q_layer(q, range(n), mats)
        _apply_cnot_layer(q, range(0, n - 1, 2), range(1, n, 2))


def run_supremacy(q, n, rng, depth):
    q.set_permutation(0)
    d = depth or n
    sq = ["sqrt_x", "s", "h"]
    for layer in range(d):
        mats = [_M1Q[sq[rng.integers(3)]] for _ in range(n)]
        _apply_1q_layer(q, range(n), mats)
        start = layer % 2
        thetas, phis, q1s, q2s = [], [], [], []
        for i in range(start, n - 1, 2):
            th, ph = rng.uniform(0, 2 * np.pi, 2)
            thetas.append(float(th))
            phis.append(float(ph))
            q1s.append(i)
            q2s.append(i + 1)
        if hasattr(q, "fsim_batch"):
            q.fsim_batch(thetas, phis, q1s, q2s)
        else:
            for th, ph, a, b in zip(thetas, phis, q1s, q2s):
                q.fsim(th, ph, a, b)


def _haar_su4(rng):
    z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
    qm, r = np.linalg.qr(z)
    return qm * (np.diag(r) / np.abs(np.diag(r)))


def run_qv(q, n, rng, depth):
    """Quantum volume, faithful protocol: depth layers of Haar-random
    SU(4) blocks on a random qubit pairing (2018 Cross et al.). Engines and
    qunit-family stacks apply each block as ONE 4x4 (Mtrx2qBatch, in-LDS-tile
    pairs fused); other stacks fall back to a u-u-CNOT-u-u approximation.
    """
    q.set_permutation(0)
    for _ in range(depth or n):
        perm = rng.permutation(n)
        pairs = [(int(perm[k]), int(perm[k + 1])) for k in range(0, n - 1, 2)]
        us = [_haar_su4(rng) for _ in pairs]
        try:
            flat = [complex(x) for u in us for x in u.flatten()]
            q.mtrx_2q_batch(flat, [a for a, _ in pairs], [b for _, b in pairs])
        except (AttributeError, RuntimeError):
            # stacks without a native 4x4: SU(4)-ish approximation
            targets, mats = [], []
            for a, b in pairs:
                for t in (a, b):
                    th, ph, lm = rng.uniform(0, 2 * np.pi, 3)
                    targets.append(t)
                    mats.append(_u_mtrx(float(th), float(ph), float(lm)))
            _apply_1q_layer(q, targets, mats)
            _apply_cnot_layer(q, [a for a, _ in pairs], [b for _, b in pairs])


def run_clifford_t_nn(q, n, rng, depth):
    """Nearest-neighbor Clifford+T (the reference's test_stabilizer_t_nn
    shape): random 1q Cliffords + T gates + NN CZ/CNOT couplers. With
    sdrp/ncrp off, the run is EXACT — T gates blocked by couplers absorb
    into the tableau via the reverse T-injection gadget; fidelity 1.0 in
    the CSV certifies no rounding happened."""
    q.set_permutation(0)
    d = depth or 8
    names = ["h", "s", "x", "z"]
    for layer in range(d):
        for i in range(n):
            r = rng.integers(6)
            if r < 4:
                getattr(q, names[r])(i)
            elif r == 4:
                q.t(i)
            # r == 5: identity
        start = layer % 2
        for i in range(start, n - 1, 2):
            if rng.integers(2):
                q.cnot(i, i + 1)
            else:
                q.cz(i, i + 1)


WORKLOADS = {
    "clifford_t_nn": run_clifford_t_nn,
    "qft": run_qft,
    "qft_cosmology": run_qft_cosmology,
    "clifford": run_clifford,
    "ghz": run_ghz,
    "random_circuit": run_random_circuit,
    "supremacy": run_supremacy,
    "qv": run_qv,
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--workload", default="qft", choices=sorted(WORKLOADS))
    p.add_argument("--min-qubits", type=int, default=4)
    p.add_argument("--max-qubits", type=int, default=24)
    p.add_argument("--samples", type=int, default=10)
    p.add_argument("--benchmark-depth", type=int, default=0)
    p.add_argument("--benchmark-shots", type=int, default=1)
    p.add_argument("--layers", default="auto",
        help="comma layer list (e.g. qunit,stabilizer_hybrid,hip) or auto")
    p.add_argument("--precision", default="fp32", choices=["fp32", "fp64"])
    p.add_argument("--seed", type=int, default=0, help="0 = time-seeded (printed for repro)")
    p.add_argument("--sdrp", type=float, default=0.0,
                   help="Schmidt-decomposition rounding parameter (0 = exact)")
    p.add_argument("--ncrp", type=float, default=0.0,
                   help="near-Clifford rounding parameter (0 = exact)")
    p.add_argument("--out", default="")
    args = p.parse_args()

    seed = args.seed or int(time.time())
    print(f"# workload={args.workload} layers={args.layers} precision={args.precision} "
          f"seed={seed}", flush=True)

    rows = ["width,samples,avg_ms,std_ms,min_ms,max_ms,fidelity"]
    fn = WORKLOADS[args.workload]
    for n in range(args.min_qubits, args.max_qubits + 1):
        if args.layers == "auto":
            q = qa.create_simulator(n, precision=args.precision, seed=seed + n)
        else:
            q = qa.create_simulator(
                n, precision=args.precision, layers=args.layers.split(","), seed=seed + n)
        if args.sdrp > 0.0:
            q.set_sdrp(args.sdrp)
        if args.ncrp > 0.0:
            q.set_ncrp(args.ncrp)
        rng = np.random.default_rng(seed + n)
        times = []
        for s in range(args.samples):
            t0 = time.perf_counter()
            fn(q, n, rng, args.benchmark_depth)
            q.multi_shot_measure_mask(
                [1 << i for i in range(min(n, 32))], args.benchmark_shots)
            if hasattr(q, "finish"):
                q.finish()
            times.append(1000.0 * (time.perf_counter() - t0))
        fid = q.get_unitary_fidelity() if hasattr(q, "get_unitary_fidelity") else 1.0
        row = (f"{n},{args.samples},{np.mean(times):.3f},{np.std(times):.3f},"
               f"{np.min(times):.3f},{np.max(times):.3f},{fid:.6f}")
        rows.append(row)
        print(row, flush=True)
    if args.out:
        with open(args.out, "w") as f:
            f.write("\n".join(rows) + "\n")


if __name__ == "__main__":
    main()
