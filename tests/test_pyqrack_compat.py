"""PyQrack-compatible wrapper: reference-user code runs unchanged
(parity: pyqrack QrackSimulator over pinvoke_api.hpp)."""

import math

import numpy as np
import pytest

from qrack_amd.pyqrack_compat import QrackSimulator


def test_bell_and_shots():
    sim = QrackSimulator(4)
    sim.h(0)
    sim.mcx([0], 1)
    assert abs(sim.prob(1) - 0.5) < 1e-5
    shots = sim.measure_shots([0, 1], 200)
    assert len(shots) == 200
    assert set(shots) <= {0, 3}


def test_gates_and_u():
    sim = QrackSimulator(2)
    sim.u(0, 0.7, 0.0, 0.4)  # phi=0: RY-like on |0>
    sim.mtrx([0, 1, 1, 0], 1)  # X
    assert abs(sim.prob(1) - 1.0) < 1e-5
    assert abs(sim.prob(0) - math.sin(0.35) ** 2) < 1e-5
    sim.r(3, -0.7, 0)  # RY(-0.7) undoes the theta rotation
    assert sim.prob(0) < 1e-5


def test_mc_family_and_fsim():
    sim = QrackSimulator(3)
    sim.x(0)
    sim.x(1)
    sim.mct([0], 1)
    sim.mcz([0, 1], 2)  # no-op on |0> target amplitude-wise
    sim.mch([1], 2)
    assert abs(sim.prob(2) - 0.5) < 1e-5
    sim.fsim(0.3, 0.2, 0, 2)
    sim.swap(0, 2)
    sim.iswap(1, 2)
    sim.adjiswap(1, 2)
    assert sim.get_error() == 0


def test_qft_lists_contiguous_and_not():
    # contiguous fast path == scattered gate path
    n = 4
    a = QrackSimulator(n)
    b = QrackSimulator(n)
    for s in (a, b):
        s.sim.set_permutation(5)
    a.qft([0, 1, 2, 3])
    b._qft_gates([0, 1, 2, 3], inverse=False)
    sva = np.asarray(a.sim.get_state_vector())
    svb = np.asarray(b.sim.get_state_vector())
    assert abs(abs(np.vdot(sva, svb)) - 1.0) < 1e-4
    a.iqft([0, 1, 2, 3])
    b._qft_gates([0, 1, 2, 3], inverse=True)
    sva = np.asarray(a.sim.get_state_vector())
    assert abs(abs(sva[5]) - 1.0) < 1e-4


def test_alu_lists():
    sim = QrackSimulator(9)
    sim.add(5, [0, 1, 2, 3])
    sim.add(11, [0, 1, 2, 3])
    assert sim.m_all() & 15 == 0  # 16 mod 16
    sim2 = QrackSimulator(9)
    sim2.add(3, [0, 1, 2])
    sim2.muln(3, 8, [0, 1, 2], [3, 4, 5])
    assert (sim2.m_all() >> 3) & 7 == 1  # 9 mod 8


def test_separability_and_knobs():
    sim = QrackSimulator(4)
    sim.h(0)
    sim.mcx([0], 1)
    assert not sim.try_separate_1qb(0)
    sim.set_sdrp(0.1)
    sim.set_ncrp(0.1)
    sim.set_reactive_separate(True)
    assert sim.get_unitary_fidelity() <= 1.0
    sim.reset_unitary_fidelity()


def test_joint_ensemble_and_expectation():
    sim = QrackSimulator(2)
    sim.h(0)
    sim.mcx([0], 1)
    # <Z Z> = +1 on Bell: probability of -1 outcome is 0
    assert abs(sim.joint_ensemble_probability([2, 2], [0, 1])) < 1e-5
    assert abs(sim.permutation_expectation([0, 1]) - 1.5) < 1e-4


def test_clone_and_ket():
    sim = QrackSimulator(3)
    sim.h(0)
    c = sim.clone()
    c.h(0)
    assert abs(c.prob(0)) < 1e-5
    assert abs(sim.prob(0) - 0.5) < 1e-5
    ket = sim.out_ket()
    sim2 = QrackSimulator(3)
    sim2.in_ket(ket)
    assert abs(sim2.prob(0) - 0.5) < 1e-5


def test_compose_and_phase_parity():
    a = QrackSimulator(2)
    b = QrackSimulator(1)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 3
    assert abs(a.prob(2) - 1.0) < 1e-5
    a.h(0)
    a.phase_parity(math.pi, [0, 2])
    a.h(0)
    assert abs(a.prob(0) - 1.0) < 1e-4  # Z from parity phase flip
