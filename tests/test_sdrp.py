"""SDRP (Schmidt-decomposition rounding parameter) — the reference's
approximate-simulation knob (pinvoke SetSdrp; qunit.cpp TrySeparate with
separabilityThreshold): sdrp > 0 rounds near-separable qubits to product
states after entangling gates, logging fidelity in GetUnitaryFidelity.
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def test_sdrp_zero_is_exact():
    q = qa.create_simulator(4, layers=["qunit", "cpu"], seed=1)
    assert q.get_sdrp() == 0.0
    q.h(0)
    q.cnot(0, 1)
    assert q.get_unitary_fidelity() == pytest.approx(1.0)
    assert abs(q.prob(1) - 0.5) < 1e-6


def test_sdrp_rounds_weakly_entangled_pairs():
    # tiny entangling rotations: sdrp rounds them away, fidelity < 1 but high
    q = qa.create_simulator(6, layers=["qunit", "cpu"], seed=2)
    q.set_sdrp(0.2)
    assert q.get_sdrp() == pytest.approx(0.2)
    for i in range(6):
        q.ry(0.3 + 0.05 * i, i)
    for d in range(3):
        for i in range(5):
            q.mcmtrx([i], [1, 0, 0, np.exp(0.05j)], i + 1)  # near-identity CPhase
    f = q.get_unitary_fidelity()
    assert 0.5 < f <= 1.0 + 1e-9
    # probabilities stay near the separable product values
    for i in range(6):
        expect = np.sin((0.3 + 0.05 * i) / 2) ** 2
        assert abs(q.prob(i) - expect) < 0.05


def test_sdrp_large_keeps_units_separable():
    # with aggressive rounding, a brickwork circuit must never grow a unit to
    # the full register (that is the entire point of the knob)
    n = 10
    q = qa.create_simulator(n, layers=["qunit", "cpu"], seed=3)
    q.set_sdrp(0.5)
    rng = np.random.default_rng(5)
    for layer in range(6):
        for i in range(n):
            q.ry(float(rng.uniform(0, 0.4)), i)
        for i in range(layer % 2, n - 1, 2):
            q.cz(i, i + 1)
    f = q.get_unitary_fidelity()
    assert 0.0 < f <= 1.0 + 1e-9
    # still usable: measurement works and norms are sane
    res = q.multi_shot_measure_mask([1 << i for i in range(n)], 100)
    assert sum(res.values()) == 100


def test_sdrp_env_knob(monkeypatch):
    monkeypatch.setenv("QRACK_QUNIT_SDRP", "0.3")
    q = qa.create_simulator(4, layers=["qunit", "cpu"], seed=4)
    assert q.get_sdrp() == pytest.approx(0.3)


def test_sdrp_exact_when_fully_entangling():
    # maximally entangled Bell pair must NOT be rounded even at high sdrp
    # (r = 0 Bloch vector: tomography refuses, state stays exact)
    q = qa.create_simulator(2, layers=["qunit", "cpu"], seed=5)
    q.set_sdrp(0.3)
    q.h(0)
    q.cnot(0, 1)
    sv = np.asarray(q.get_state_vector())
    s = 1 / np.sqrt(2)
    assert_states_close(sv, np.array([s, 0, 0, s]), 1e-5)
    assert q.get_unitary_fidelity() == pytest.approx(1.0)


# ---- NCRP (near-Clifford rounding parameter) --------------------------------


def test_ncrp_zero_switches_to_engine():
    # a T gate then CNOT stays Clifford via the reverse T-injection gadget
    # (round 2); with the gadget disabled it must force the dense engine
    import os

    os.environ["QRACK_USE_T_GADGET"] = "0"
    try:
        q = qa.create_simulator(4, layers=["stabilizer_hybrid", "cpu"], seed=1)
        q.h(0)
        q.t(0)
        q.cnot(0, 1)
        assert not q.is_clifford()
    finally:
        del os.environ["QRACK_USE_T_GADGET"]
    q2 = qa.create_simulator(4, layers=["stabilizer_hybrid", "cpu"], seed=1)
    q2.h(0)
    q2.t(0)
    q2.cnot(0, 1)
    assert q2.is_clifford()  # gadget kept the tableau


def test_ncrp_rounds_near_clifford_phases():
    # rz(0.1) is within ncrp=0.1 of identity: the tableau absorbs it and the
    # hybrid never materializes a state vector
    q = qa.create_simulator(30, layers=["stabilizer_hybrid", "sparse"], seed=2)
    q.set_ncrp(0.1)
    assert q.get_ncrp() == pytest.approx(0.1)
    for i in range(30):
        q.h(i)
    for i in range(29):
        q.rz(0.08, i)
        q.cnot(i, i + 1)
    assert q.is_clifford()  # still a tableau at width 30
    f = q.get_unitary_fidelity()
    assert 0.5 < f < 1.0  # rounding happened and was logged
    res = q.multi_shot_measure_mask([1, 2, 4], 50)
    assert sum(res.values()) == 50


def test_ncrp_keeps_true_t_gates_exact():
    # T (pi/4) is NOT within ncrp=0.1 of Clifford: must stay exact
    q = qa.create_simulator(3, layers=["stabilizer_hybrid", "cpu"], seed=3)
    q.set_ncrp(0.1)
    cp = qa.create_simulator(3, engine="cpu", seed=3)
    for s in (q, cp):
        s.h(0)
        s.t(0)
        s.cnot(0, 1)
        s.h(1)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)
    assert q.get_unitary_fidelity() == pytest.approx(1.0)


def test_ncrp_snap_accuracy():
    # rounding rz(delta) ~ identity: state equals the unrounded Clifford part
    q = qa.create_simulator(2, layers=["stabilizer_hybrid", "cpu"], seed=4)
    q.set_ncrp(0.05)
    q.h(0)
    q.rz(0.06, 0)  # |sin(0.03)| = 0.03 <= 0.05: rounded away
    q.cnot(0, 1)
    cp = qa.create_simulator(2, engine="cpu", seed=4)
    cp.h(0)
    cp.cnot(0, 1)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)
    # fidelity log matches 1 - 2 p (1-p) (1 - cos 0.06) with p = 0.5
    expect = 1.0 - 2.0 * 0.25 * (1.0 - np.cos(0.06))
    assert q.get_unitary_fidelity() == pytest.approx(expect, abs=1e-6)


def test_ncrp_through_qunit_stack():
    # QUnit forwards ncrp to its (current and future) units and aggregates
    # the units' fidelities
    q = qa.create_simulator(8, layers=["qunit", "stabilizer_hybrid", "cpu"], seed=5)
    q.set_ncrp(0.1)
    for i in range(8):
        q.h(i)
        q.rz(0.05, i)
    for i in range(7):
        q.cnot(i, i + 1)
    f = q.get_unitary_fidelity()
    assert 0.8 < f < 1.0
    q.reset_unitary_fidelity()
    assert q.get_unitary_fidelity() == pytest.approx(1.0)


def test_reactive_separate_mirror_circuit():
    # with reactive separation, a mirror circuit returns every qubit to its
    # own 1-qubit unit and the state is exact
    q = qa.create_simulator(6, layers=["qunit", "cpu"], seed=6)
    q.set_reactive_separate(True)
    assert q.get_reactive_separate()
    ops = []
    rng = np.random.default_rng(8)
    for _ in range(10):
        a, b = rng.choice(6, 2, replace=False)
        th = float(rng.uniform(0, 2 * np.pi))
        ops.append((int(a), int(b), th))
    for a, b, th in ops:
        q.ry(th, a)
        q.cnot(a, b)
    for a, b, th in reversed(ops):
        q.cnot(a, b)
        q.ry(-th, a)
    for i in range(6):
        assert q.prob(i) < 1e-4
    assert q.get_unitary_fidelity() == pytest.approx(1.0, abs=1e-6)


def test_stochastic_near_clifford():
    """Stochastic near-Clifford rounding (QRACK_USE_APPROX_NEAR_CLIFFORD /
    set_stochastic): blocked T shards snap stochastically to Clifford —
    the tableau never materializes, fidelity < 1 is reported, and the
    SHOT AVERAGE of an H-T-H interference stays near the exact value."""
    import numpy as np

    exact = np.sin(np.pi / 8) ** 2  # P(1) for H T H
    ones = 0
    shots = 1500
    for s in range(shots):
        q = qa.create_simulator(2, layers=["stabilizer_hybrid", "cpu"], seed=1000 + s)
        q.set_stochastic(True)
        q.h(0)
        q.t(0)
        q.cnot(0, 1)  # blocks the shard -> stochastic snap (no ancilla)
        q.cnot(0, 1)
        q.h(0)
        assert q.is_clifford()
        assert q.ancilla_count() == 0
        ones += q.m(0)
    p1 = ones / shots
    # the stochastic S-or-identity snap is NOT unbiased for interference
    # terms (reference's caveat); accept a broad band around the exact
    # value but far from 0/0.5 degeneracy
    assert 0.05 < p1 < 0.35, p1


def test_fidelity_guard_throws_and_opts_out():
    """Reference CheckFidelity parity: deep ACE-elided circuits that drive
    the fidelity estimate to ~0 raise; QRACK_DISABLE_QUNIT_FIDELITY_GUARD
    opts out."""
    import os

    import numpy as np

    def grind(n_rounds):
        os.environ["QRACK_QUNIT_ACE_MAX_QB"] = "2"
        try:
            q = qa.create_simulator(6, layers=["qunit", "cpu"], seed=5)
            rng = np.random.default_rng(5)
            for _ in range(n_rounds):
                for t in range(6):
                    q.ry(float(rng.uniform(0.5, 1.0)), t)
                for a in range(5):
                    q.cnot(a, a + 1)  # every coupler elides under the cap
            return q.get_unitary_fidelity()
        finally:
            del os.environ["QRACK_QUNIT_ACE_MAX_QB"]

    with pytest.raises(Exception):
        grind(60)
    os.environ["QRACK_DISABLE_QUNIT_FIDELITY_GUARD"] = "1"
    try:
        f = grind(60)
        assert 0.0 <= f < 1e-10
    finally:
        del os.environ["QRACK_DISABLE_QUNIT_FIDELITY_GUARD"]
