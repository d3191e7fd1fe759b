"""Distributed QPager tests on CPU (gloo backend, world_size=2/4).

Validates the multi-rank page protocol — meta-qubit exchanges, page
relabeling, semi-meta controls, distributed measurement — against the
single-process reference, with no GPU. The same code path drives RCCL on
MI355X (reference parity model: QPager tested over repeated single-device
pages, test_main.cpp:277-283 — N pages on 1 device == N pages on N).
"""

import multiprocessing as mp
import os

import numpy as np
import pytest

import os as _os
PORT = 20000 + (_os.getpid() % 8000)


def _run_worker(rank, world, fn_name, q, seed, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from qrack_amd.dist_pager import DistQPager

    pager = DistQPager(q, engine="cpu", seed=seed)
    result = globals()[fn_name](pager, rank)
    dist.barrier()
    dist.destroy_process_group()
    if rank == 0 and result is not None:
        q_out.put(result)


q_out = None


def run_distributed(fn_name, world=2, qubits=4, seed=7, port_off=0):
    global q_out
    ctx = mp.get_context("spawn")
    q_out = ctx.Queue()
    procs = []
    for r in range(world):
        p = ctx.Process(
            target=_worker_entry, args=(r, world, fn_name, qubits, seed, PORT + port_off, q_out)
        )
        p.start()
        procs.append(p)
    ok = True
    for p in procs:
        p.join(timeout=120)
        if p.exitcode != 0:
            ok = False
    assert ok, f"distributed workers failed for {fn_name}"


def _worker_entry(rank, world, fn_name, qubits, seed, port, out_q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from qrack_amd.dist_pager import DistQPager

    pager = DistQPager(qubits, engine="cpu", seed=seed)
    globals()[fn_name](pager, rank)
    dist.barrier()
    dist.destroy_process_group()


# ---- worker bodies (asserts run on every rank) ------------------------------


def _body_ghz(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(0)
    for i in range(n - 1):
        pager.cnot(i, i + 1)  # last cnot targets a meta qubit
    sv = pager.get_state_vector()
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-5
    assert abs(abs(sv[-1]) - 1 / np.sqrt(2)) < 1e-5
    assert np.sum(np.abs(sv) ** 2) == pytest.approx(1.0, abs=1e-5)
    assert abs(pager.prob(n - 1) - 0.5) < 1e-5


def _body_meta_h(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(n - 1)  # H directly on a meta qubit: exchange path
    sv = pager.get_state_vector()
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-5
    assert abs(abs(sv[1 << (n - 1)]) - 1 / np.sqrt(2)) < 1e-5
    pager.h(n - 1)
    sv = pager.get_state_vector()
    assert abs(abs(sv[0]) - 1.0) < 1e-5


def _body_meta_x_relabel(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.x(n - 1)  # page relabel, no comm
    sv = pager.get_state_vector()
    assert abs(abs(sv[1 << (n - 1)]) - 1.0) < 1e-6
    pager.y(n - 1)  # relabel + phase
    sv = pager.get_state_vector()
    assert abs(sv[0] - (-1j)) < 1e-5 or abs(sv[0] - 1j) < 1e-5


def _body_qft_vs_reference(pager, rank):
    n = pager.num_qubits
    x = 5
    pager.set_permutation(x)
    pager.qft(0, n)
    sv = pager.get_state_vector().astype(np.complex128)
    N = 1 << n
    k = np.arange(N)
    expected = np.exp(2j * np.pi * x * k / N) / np.sqrt(N)
    rev = np.array([int(format(i, f"0{n}b")[::-1], 2) for i in range(N)])
    got = sv[rev]
    inner = np.vdot(expected, got)
    assert abs(abs(inner) - 1.0) < 1e-4


def _body_qft_roundtrip(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(3)
    pager.qft(0, n)
    pager.iqft(0, n)
    sv = pager.get_state_vector()
    assert abs(abs(sv[3]) - 1.0) < 1e-4


def _body_compare_single_process(pager, rank):
    """Random circuit on the pager == same circuit on one CPU engine."""
    import qrack_amd as qa

    n = pager.num_qubits
    ref = qa.create_simulator(n, engine="cpu", seed=99)
    rng = np.random.default_rng(13)
    pager.set_permutation(0)
    for layer in range(4):
        for i in range(n):
            th = float(rng.uniform(0, 2 * np.pi))
            s, c = np.sin(th / 2), np.cos(th / 2)
            m = [c, -s, s, c]  # RY
            pager.mtrx(m, i)
            ref.ry(th, i)
        for i in range(n - 1):
            pager.cnot(i, i + 1)
            ref.cnot(i, i + 1)
        t = int(rng.integers(n))
        pager.t(t)
        ref.t(t)
        a, b = rng.choice(n, 2, replace=False)
        pager.cz(int(a), int(b))
        ref.cz(int(a), int(b))
    sv = pager.get_state_vector().astype(np.complex128)
    rv = np.asarray(ref.get_state_vector()).astype(np.complex128)
    inner = abs(np.vdot(rv, sv))
    assert inner > 1 - 1e-4, f"fidelity {inner}"


def _body_measure(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(0)
    pager.cnot(0, n - 1)
    res = pager.multi_shot_measure_mask([1, 1 << (n - 1)], 500)
    assert sum(res.values()) == 500
    assert set(res.keys()) <= {0, 3}
    assert 150 < res.get(0, 0) < 350
    r = pager.m_all()
    assert r in (0, (1 << (n - 1)) | 1)
    sv = pager.get_state_vector()
    assert abs(abs(sv[r]) - 1.0) < 1e-6


def _body_force_m_meta(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(0)
    pager.h(n - 1)
    out = pager.force_m(n - 1, True)
    assert out is True
    assert abs(pager.prob(n - 1) - 1.0) < 1e-5


def _body_swap_meta(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(1)  # qubit 0 set
    pager.swap(0, n - 1)
    sv = pager.get_state_vector()
    assert abs(abs(sv[1 << (n - 1)]) - 1.0) < 1e-5


# ---- tests -------------------------------------------------------------------


def test_ghz_world2():
    run_distributed("_body_ghz", world=2, qubits=4, port_off=1)


def test_meta_h_world2():
    run_distributed("_body_meta_h", world=2, qubits=4, port_off=2)


def test_meta_x_relabel_world2():
    run_distributed("_body_meta_x_relabel", world=2, qubits=4, port_off=3)


def test_qft_vs_reference_world2():
    run_distributed("_body_qft_vs_reference", world=2, qubits=5, port_off=4)


def test_qft_roundtrip_world4():
    run_distributed("_body_qft_roundtrip", world=4, qubits=6, port_off=5)


def test_compare_single_process_world2():
    run_distributed("_body_compare_single_process", world=2, qubits=5, port_off=6)


def test_compare_single_process_world4():
    run_distributed("_body_compare_single_process", world=4, qubits=6, port_off=7)


def test_measure_world2():
    run_distributed("_body_measure", world=2, qubits=4, port_off=8)


def test_force_m_meta_world2():
    run_distributed("_body_force_m_meta", world=2, qubits=4, port_off=9)


def test_swap_meta_world2():
    run_distributed("_body_swap_meta", world=2, qubits=4, port_off=10)


def _body_qft_after_swaps(pager, rank):
    # scramble the lazy map with logical swaps, then full QFT: exercises the
    # fused column kernel's scattered-bit ramps under a permuted map
    n = pager.num_qubits
    x = 3
    pager.set_permutation(x)
    pager.swap(0, n - 1)
    pager.swap(1, n - 2)
    pager.swap(0, n - 1)  # permutation is now a single (1, n-2) swap
    # logical value after swaps: bit1 <-> bit(n-2) of x=3 -> 1 | (1 << (n-2))
    xl = 1 | (1 << (n - 2))
    pager.qft(0, n)
    sv = pager.get_state_vector().astype(np.complex128)
    N = 1 << n
    k = np.arange(N)
    expected = np.exp(2j * np.pi * xl * k / N) / np.sqrt(N)
    rev = np.array([int(format(i, f"0{n}b")[::-1], 2) for i in range(N)])
    inner = np.vdot(expected, sv[rev])
    assert abs(abs(inner) - 1.0) < 1e-4


def _body_iqft_roundtrip_swapped(pager, rank):
    n = pager.num_qubits
    pager.set_permutation(9 % (1 << n))
    pager.swap(0, n - 1)
    pager.qft(0, n)
    pager.iqft(0, n)
    pager.swap(0, n - 1)
    sv = pager.get_state_vector()
    assert abs(abs(sv[9 % (1 << n)]) - 1.0) < 1e-4


def test_qft_after_swaps_world4():
    run_distributed("_body_qft_after_swaps", world=4, qubits=6, seed=11, port_off=10)


def test_iqft_roundtrip_swapped_world2():
    run_distributed("_body_iqft_roundtrip_swapped", world=2, qubits=5, seed=12, port_off=11)


def _body_deep_equivalence(pager, rank):
    """Deeper/wider equivalence sweep: many meta-ops,
    swaps and mid-circuit probability probes at 10 qubits, world up to 8."""
    import qrack_amd as qa

    n = pager.num_qubits
    ref = qa.create_simulator(n, engine="cpu", seed=99)
    rng = np.random.default_rng(29)
    pager.set_permutation(0)
    for layer in range(6):
        for i in range(n):
            th = float(rng.uniform(0, 2 * np.pi))
            s, c = np.sin(th / 2), np.cos(th / 2)
            pager.mtrx([c, -s, s, c], i)
            ref.ry(th, i)
        # meta-heavy two-qubit ops: always touch the top qubits
        for hi in range(n - 1, n - 4, -1):
            lo = int(rng.integers(n - 4))
            pager.cnot(lo, hi)
            ref.cnot(lo, hi)
        a, b = rng.choice(n, 2, replace=False)
        pager.swap(int(a), int(b))
        ref.swap(int(a), int(b))
        t = int(rng.integers(n))
        pager.rz(0.37, t)
        ref.rz(0.37, t)
        # probability probe mid-circuit on a (possibly meta) qubit
        q = int(rng.integers(n))
        assert abs(pager.prob(q) - ref.prob(q)) < 1e-5
    pager.qft(0, n)
    ref.qft(0, n)
    pager.iqft(0, n)
    ref.iqft(0, n)
    sv = pager.get_state_vector().astype(np.complex128)
    rv = np.asarray(ref.get_state_vector()).astype(np.complex128)
    inner = abs(np.vdot(rv, sv))
    assert inner > 1 - 1e-4, f"fidelity {inner}"


def test_deep_equivalence_world4():
    run_distributed("_body_deep_equivalence", world=4, qubits=10, port_off=12)


def test_deep_equivalence_world8():
    run_distributed("_body_deep_equivalence", world=8, qubits=10, port_off=13)
