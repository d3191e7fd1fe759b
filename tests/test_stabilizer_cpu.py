"""QStabilizer (CHP tableau) tests vs the dense CPU engine.

Parity model: /root/reference/test/tests.cpp stabilizer cases + the
engine-matrix rerun (same circuits on tableau and state-vector backends
must agree).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def make_stab(n, seed=7):
    return qa.create_simulator(n, layers=["stabilizer"], seed=seed)


def make_cpu(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


CLIFFORD_1Q = ["h", "x", "y", "z", "s", "is_", "sqrt_x", "isqrt_x"]
CLIFFORD_2Q = ["cnot", "cz", "cy", "swap", "iswap"]


@pytest.mark.parametrize("seed", [1, 2, 3, 4, 5])
def test_random_clifford_circuit_vs_dense(seed):
    n = 6
    rng = np.random.default_rng(seed)
    st = make_stab(n, seed=seed)
    cp = make_cpu(n, seed=seed)
    for _ in range(40):
        if rng.random() < 0.6:
            g = CLIFFORD_1Q[rng.integers(len(CLIFFORD_1Q))]
            t = int(rng.integers(n))
            getattr(st, g)(t)
            getattr(cp, g)(t)
        else:
            g = CLIFFORD_2Q[rng.integers(len(CLIFFORD_2Q))]
            a, b = rng.choice(n, 2, replace=False)
            getattr(st, g)(int(a), int(b))
            getattr(cp, g)(int(a), int(b))
    sv_st = st.get_state_vector()
    sv_cp = cp.get_state_vector()
    assert_states_close(sv_st, sv_cp, 1e-5)


def test_bell_and_ghz():
    st = make_stab(3)
    st.h(0)
    st.cnot(0, 1)
    st.cnot(1, 2)
    sv = st.get_state_vector()
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-6
    assert abs(abs(sv[7]) - 1 / np.sqrt(2)) < 1e-6
    assert abs(st.prob(2) - 0.5) < 1e-6


def test_prob_deterministic():
    st = make_stab(2)
    st.x(0)
    assert st.prob(0) == 1.0
    assert st.prob(1) == 0.0
    st.h(1)
    assert st.prob(1) == 0.5


def test_measurement_collapse():
    st = make_stab(2, seed=5)
    st.h(0)
    st.cnot(0, 1)
    r0 = st.m(0)
    r1 = st.m(1)
    assert r0 == r1
    # deterministic after collapse
    assert st.prob(0) in (0.0, 1.0)


def test_force_m():
    st = make_stab(2, seed=5)
    st.h(0)
    st.cnot(0, 1)
    r = st.force_m(0, True)
    assert r is True
    assert st.prob(1) == 1.0


def test_measurement_statistics():
    ones = 0
    for seed in range(100):
        st = make_stab(1, seed=seed)
        st.h(0)
        if st.m(0):
            ones += 1
    assert 25 < ones < 75


def test_phase_gate_global_phase_tracked():
    # S|1> = i|1>
    st = make_stab(1)
    st.x(0)
    st.s(0)
    sv = st.get_state_vector()
    assert abs(sv[1] - 1j) < 1e-6


def test_mall_and_multishot():
    st = make_stab(3, seed=9)
    st.h(0)
    st.cnot(0, 1)
    res = st.multi_shot_measure_mask([1, 2], 400)
    assert sum(res.values()) == 400
    assert set(res.keys()) <= {0, 3}
    assert 120 < res.get(0, 0) < 280
    r = st.m_all()
    assert r in (0, 3)


def test_compose():
    a = make_stab(1, seed=1)
    a.h(0)
    b = make_stab(1, seed=2)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 2
    sv = a.get_state_vector()
    assert abs(abs(sv[2]) - 1 / np.sqrt(2)) < 1e-6
    assert abs(abs(sv[3]) - 1 / np.sqrt(2)) < 1e-6


def test_compose_entangled():
    a = make_stab(1, seed=1)
    a.h(0)
    b = make_stab(2, seed=2)
    b.h(0)
    b.cnot(0, 1)
    a.compose(b)
    assert a.num_qubits == 3
    cp = make_cpu(3)
    cp.h(0)
    cp.h(1)
    cp.cnot(1, 2)
    assert_states_close(a.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_dispose_separable():
    st = make_stab(3, seed=3)
    st.h(0)
    st.x(1)
    st.h(2)
    st.z(2)
    st.dispose(1, 1)
    assert st.num_qubits == 2
    cp = make_cpu(2)
    cp.h(0)
    cp.h(1)
    cp.z(1)
    assert_states_close(st.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_dispose_entangled_throws():
    st = make_stab(2, seed=3)
    st.h(0)
    st.cnot(0, 1)
    with pytest.raises(Exception):
        st.dispose(0, 1)


def test_decompose():
    st = make_stab(3, seed=4)
    st.h(1)
    st.cnot(1, 2)  # (1,2) Bell, 0 free
    st.h(0)
    dest = make_stab(2, seed=5)
    st.decompose(1, dest)
    assert st.num_qubits == 1
    assert dest.num_qubits == 2
    sv = dest.get_state_vector()
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-6
    assert abs(abs(sv[3]) - 1 / np.sqrt(2)) < 1e-6
    assert abs(st.prob(0) - 0.5) < 1e-6


def test_allocate():
    st = make_stab(2, seed=6)
    st.h(0)
    st.allocate(2)
    assert st.num_qubits == 4
    assert st.prob(2) == 0.0
    assert st.prob(3) == 0.0


def test_clone_and_compare():
    st = make_stab(4, seed=7)
    st.h(0)
    st.cnot(0, 3)
    c = st.clone()
    assert st.approx_compare(c)
    c.x(1)
    assert not st.approx_compare(c)


def test_mirror_clifford():
    n = 5
    rng = np.random.default_rng(11)
    st = make_stab(n, seed=11)
    ops = []
    for _ in range(30):
        if rng.random() < 0.6:
            g = ["h", "s", "x", "z"][rng.integers(4)]
            t = int(rng.integers(n))
            getattr(st, g)(t)
            ops.append((g, t))
        else:
            a, b = rng.choice(n, 2, replace=False)
            st.cnot(int(a), int(b))
            ops.append(("cnot", int(a), int(b)))
    inv = {"h": "h", "s": "is_", "x": "x", "z": "z"}
    for op in reversed(ops):
        if op[0] == "cnot":
            st.cnot(op[1], op[2])
        else:
            getattr(st, inv[op[0]])(op[1])
    assert st.m_all() == 0


def test_non_clifford_throws():
    st = make_stab(2)
    with pytest.raises(Exception):
        st.t(0)
    with pytest.raises(Exception):
        st.rx(0.3, 0)
    with pytest.raises(Exception):
        st.ccnot(0, 1, 1)


def test_amplitude_access():
    st = make_stab(3, seed=2)
    st.h(0)
    st.cnot(0, 2)
    a0 = st.get_amplitude(0)
    a5 = st.get_amplitude(5)
    a1 = st.get_amplitude(1)
    assert abs(abs(a0) - 1 / np.sqrt(2)) < 1e-6
    assert abs(abs(a5) - 1 / np.sqrt(2)) < 1e-6
    assert abs(a1) < 1e-9
