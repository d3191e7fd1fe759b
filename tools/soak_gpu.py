"""Sustained-load GPU soak: repeated create/run/destroy cycles across the
layer stacks with VRAM watermark tracking — catches leaks and instability
that single-shot benchmarks miss. Run on an MI355X box:

    python tools/soak_gpu.py --cycles 30 --qubits 26
"""

import argparse
import subprocess
import sys
import time

sys.path.insert(0, ".")

import numpy as np

import qrack_amd as qa


def vram_used_mb():
    try:
        out = subprocess.run(["rocm-smi", "--showmeminfo", "vram"], capture_output=True,
                             text=True, timeout=30).stdout
        for line in out.splitlines():
            if "Used Memory" in line:
                return int(line.split(":")[-1].strip()) // (1024 * 1024)
    except Exception:
        pass
    return -1


def cycle(i, n):
    rng = np.random.default_rng(i)
    # dense engine circuit
    q = qa.create_simulator(n, engine="hip", seed=i)
    q.set_permutation(int(rng.integers(1 << min(n, 62))))
    q.qft(0, n)
    q.multi_shot_measure_mask([1 << b for b in range(min(n, 16))], 4)
    del q
    # layered stack circuit
    s = qa.create_simulator(min(n + 6, 36), layers=["qunit", "stabilizer_hybrid", "hip"], seed=i)
    for b in range(12):
        s.h(b)
    for b in range(11):
        s.cnot(b, b + 1)
    s.t(int(rng.integers(12)))
    s.m_all()
    del s
    # fuser stack
    f = qa.create_simulator(n - 2, layers=["fuser", "hip"], seed=i)
    for d in range(4):
        for b in range(n - 2):
            f.ry(0.1 + 0.01 * b, b)
        for b in range(0, n - 3, 2):
            f.cnot(b, b + 1)
    f.finish()
    del f


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cycles", type=int, default=30)
    p.add_argument("--qubits", type=int, default=26)
    args = p.parse_args()
    base = vram_used_mb()
    t0 = time.perf_counter()
    peak = base
    for i in range(args.cycles):
        cycle(i, args.qubits)
        used = vram_used_mb()
        peak = max(peak, used)
        if (i + 1) % 10 == 0:
            print(f"cycle {i + 1}/{args.cycles}: vram {used} MB (base {base})", flush=True)
    t1 = time.perf_counter()
    final = vram_used_mb()
    print(f"SOAK done: {args.cycles} cycles in {t1 - t0:.1f}s; "
          f"vram base={base} final={final} peak={peak} MB")
    # allow small allocator pools; a leak would grow with cycles
    if base > 0 and final > base + 2048:
        print("SOAK FAIL: vram grew > 2 GB")
        sys.exit(1)
    print("SOAK OK")


if __name__ == "__main__":
    main()
