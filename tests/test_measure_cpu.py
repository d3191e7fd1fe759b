"""Measurement / probability / sampling tests (CPU engine).

Parity model: /root/reference/test/tests.cpp measurement cases
(test_m, test_multishot, prob family).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import RefSim


def make(n, seed=7, precision="fp32"):
    return qa.create_simulator(n, precision=precision, engine="cpu", seed=seed)


def test_prob_basics():
    q = make(2)
    q.h(0)
    assert abs(q.prob(0) - 0.5) < 1e-6
    assert abs(q.prob(1)) < 1e-6
    q.cnot(0, 1)
    assert abs(q.prob(1) - 0.5) < 1e-6


def test_prob_mask_reg():
    q = make(3)
    q.h(0)
    q.cnot(0, 1)
    # state (|000> + |011>)/sqrt2
    assert abs(q.prob_mask(0b011, 0b011) - 0.5) < 1e-6
    assert abs(q.prob_mask(0b011, 0b001)) < 1e-6
    assert abs(q.prob_reg(0, 2, 0b11) - 0.5) < 1e-6
    assert abs(q.prob_parity(0b011)) < 1e-6
    assert abs(q.prob_parity(0b001) - 0.5) < 1e-6


def test_force_m_collapse():
    q = make(2, seed=5)
    q.h(0)
    q.cnot(0, 1)
    r = q.force_m(0, True)
    assert r is True
    assert abs(q.prob(1) - 1.0) < 1e-6
    sv = q.get_state_vector()
    assert abs(abs(sv[3]) - 1.0) < 1e-6


def test_m_all_distribution():
    counts = {0: 0, 3: 0}
    for seed in range(200):
        q = make(2, seed=seed)
        q.h(0)
        q.cnot(0, 1)
        r = q.m_all()
        assert r in (0, 3)
        counts[r] += 1
    assert 50 < counts[0] < 150


def test_multi_shot_measure_mask():
    q = make(3, seed=9)
    q.h(0)
    q.cnot(0, 1)
    shots = 2000
    res = q.multi_shot_measure_mask([1, 2], shots)
    assert sum(res.values()) == shots
    assert set(res.keys()) <= {0, 3}
    assert 800 < res.get(0, 0) < 1200


def test_multi_shot_does_not_collapse():
    q = make(2, seed=9)
    q.h(0)
    q.multi_shot_measure_mask([1], 100)
    assert abs(q.prob(0) - 0.5) < 1e-6


def test_expectation_variance():
    q = make(3, seed=2)
    q.x(1)  # |010> = 2
    assert abs(q.expectation_bits_all([0, 1, 2]) - 2.0) < 1e-6
    assert abs(q.variance_bits_all([0, 1, 2])) < 1e-6
    q.h(0)  # (|010>+|011>)/sqrt2 -> values 2,3
    assert abs(q.expectation_bits_all([0, 1, 2]) - 2.5) < 1e-5
    assert abs(q.variance_bits_all([0, 1, 2]) - 0.25) < 1e-5


def test_pauli_expectation():
    q = make(2, seed=2)
    q.h(0)
    # <X> on |+> = 1
    assert abs(q.pauli_expectation([0], [qa.Pauli_X]) - 1.0) < 1e-5
    # <Z> on |+> = 0
    assert abs(q.pauli_expectation([0], [qa.Pauli_Z])) < 1e-5
    # Bell: <ZZ> = 1, <XX> = 1
    q2 = make(2, seed=3)
    q2.h(0)
    q2.cnot(0, 1)
    assert abs(q2.pauli_expectation([0, 1], [qa.Pauli_Z, qa.Pauli_Z]) - 1.0) < 1e-5
    assert abs(q2.pauli_expectation([0, 1], [qa.Pauli_X, qa.Pauli_X]) - 1.0) < 1e-5


def test_force_m_parity():
    q = make(2, seed=4)
    q.h(0)
    q.cnot(0, 1)  # parity of (0,1) is 0 always
    r = q.force_m_parity(0b11, False, do_force=False)
    assert r is False
    q.h(0)
    r2 = q.force_m_parity(0b01, True)
    assert r2 is True
    assert abs(q.prob(0) - 1.0) < 1e-5


def test_seeded_determinism():
    r1 = []
    for _ in range(2):
        q = make(4, seed=123)
        for i in range(4):
            q.h(i)
        r1.append(q.m_all())
    assert r1[0] == r1[1]


def test_reduced_density_matrix():
    import numpy as np

    q = make(2, seed=3)
    q.h(0)
    rho = np.asarray(q.reduced_density_matrix(0))
    assert abs(rho[0, 0] - 0.5) < 1e-5
    assert abs(rho[0, 1] - 0.5) < 1e-5  # coherent |+>
    q.cnot(0, 1)  # now maximally mixed RDM
    rho = np.asarray(q.reduced_density_matrix(0))
    assert abs(rho[0, 0] - 0.5) < 1e-5
    assert abs(rho[0, 1]) < 1e-5
    # QUnit fast path
    qu = qa.create_simulator(3, layers=["qunit", "cpu"], seed=4)
    qu.h(1)
    rho = np.asarray(qu.reduced_density_matrix(1))
    assert abs(rho[0, 1] - 0.5) < 1e-5
