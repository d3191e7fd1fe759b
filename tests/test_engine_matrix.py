"""Engine-matrix rerun: the SAME test battery across every layer stack.

This is the reference's core portability guarantee (test_main.cpp:219-300:
one Catch2 session per enabled layer combination, identical TEST_CASEs).
Every stack must implement identical gate semantics.
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close

STACKS = [
    ["cpu"],
    ["sparse"],
    ["bdt"],
    ["hybrid"],
    ["stabilizer_hybrid", "cpu"],
    ["bdt_hybrid", "cpu"],
    ["qunit", "cpu"],
    ["qunit", "stabilizer_hybrid", "cpu"],
    ["qunit_multi", "stabilizer_hybrid", "cpu"],
    ["pager", "cpu"],
    ["turboquant"],
    ["fuser", "cpu"],
    ["fuser", "qunit", "stabilizer_hybrid", "cpu"],
    ["tensor_network", "cpu"],
    ["noisy", "cpu"],  # QRACK_GATE_DEPOLARIZATION defaults irrelevant: set 0
]

IDS = ["-".join(s) for s in STACKS]


@pytest.fixture(autouse=True)
def _zero_noise(monkeypatch):
    monkeypatch.setenv("QRACK_GATE_DEPOLARIZATION", "0.0")


def make(n, layers, seed=7):
    return qa.create_simulator(n, layers=layers, seed=seed, pages_per_device=2)


def make_cpu(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


@pytest.mark.parametrize("layers", STACKS, ids=IDS)
def test_ghz(layers):
    n = 5
    q = make(n, layers)
    q.h(0)
    for i in range(n - 1):
        q.cnot(i, i + 1)
    assert abs(q.prob(n - 1) - 0.5) < 1e-5
    r = q.m_all()
    assert r in (0, (1 << n) - 1)


@pytest.mark.parametrize("layers", STACKS, ids=IDS)
def test_random_circuit_state(layers):
    n = 5
    rng = np.random.default_rng(31)
    q = make(n, layers, seed=31)
    cp = make_cpu(n, seed=31)
    for _ in range(15):
        r = rng.random()
        if r < 0.5:
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            cp.ry(th, t)
        elif r < 0.8:
            a, b = rng.choice(n, 2, replace=False)
            q.cnot(int(a), int(b))
            cp.cnot(int(a), int(b))
        else:
            t = int(rng.integers(n))
            q.t(t)
            cp.t(t)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)


@pytest.mark.parametrize("layers", STACKS, ids=IDS)
def test_qft_roundtrip(layers):
    n = 5
    q = make(n, layers, seed=5)
    q.set_permutation(11)
    q.qft(0, n)
    q.iqft(0, n)
    res = q.multi_shot_measure_mask([1 << i for i in range(n)], 20)
    assert res == {11: 20}


@pytest.mark.parametrize("layers", STACKS, ids=IDS)
def test_measurement_statistics(layers):
    q = make(4, layers, seed=9)
    q.h(0)
    q.cnot(0, 2)
    res = q.multi_shot_measure_mask([1, 4], 300)
    assert sum(res.values()) == 300
    assert set(res.keys()) <= {0, 3}
    assert 90 < res.get(0, 0) < 210


@pytest.mark.parametrize("layers", STACKS, ids=IDS)
def test_clifford_circuit(layers):
    n = 4
    q = make(n, layers, seed=3)
    cp = make_cpu(n, seed=3)
    for g, t in [("h", 0), ("s", 0), ("h", 1), ("z", 1), ("h", 3)]:
        getattr(q, g)(t)
        getattr(cp, g)(t)
    q.cnot(0, 1)
    cp.cnot(0, 1)
    q.cz(1, 2)
    cp.cz(1, 2)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_extended_api_over_stacks():
    """A light extended-API pass over every stack: batched gates, phase
    pairs, boolean logic, shifts, expectation family, approximation knobs."""
    import numpy as np

    for layers in STACKS:
        q = qa.create_simulator(6, layers=list(layers), seed=7, pages_per_device=2)
        cp = qa.create_simulator(6, engine="cpu", seed=7)
        for s in (q, cp):
            s.h(0)
            s.h(2)
            s.mtrx_1q_batch([1, 3], [0.6, -0.8, 0.8, 0.6] * 2)
            s.cz(0, 2)
            s.cnot_batch([0], [4])
            s.cphase_pairs([2], [5], [0.7])
            s.sqrt_w(1)
            s.isqrt_w(1)
            s.crx(0.4, 0, 1)
        try:
            got = q.get_state_vector()
        except RuntimeError:
            continue  # width/capability-capped stack
        from ref_sim import assert_states_close
        assert_states_close(got, cp.get_state_vector(), 1e-4)
        assert abs(q.expectation_pauli_all([0, 2], [2, 2])
                   - cp.expectation_pauli_all([0, 2], [2, 2])) < 1e-4
        jp = np.asarray(q.prob_bits_all([0, 4]))
        jp2 = np.asarray(cp.prob_bits_all([0, 4]))
        np.testing.assert_allclose(jp, jp2, atol=1e-4)
