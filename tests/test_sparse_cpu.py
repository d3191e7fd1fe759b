"""Sparse state-vector engine tests (parity: StateVectorSparse,
statevector.hpp:248-310): same gate semantics as dense, O(support) cost."""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def make_sp(n, seed=7):
    return qa.create_simulator(n, layers=["sparse"], seed=seed)


def make_cpu(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_random_circuit_vs_dense(seed):
    n = 6
    rng = np.random.default_rng(seed)
    q = make_sp(n, seed=seed)
    cp = make_cpu(n, seed=seed)
    for _ in range(20):
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


def test_wide_sparse_state():
    # 40 qubits dense would be 16 TB; sparse keeps a handful of amplitudes
    n = 40
    q = make_sp(n, seed=2)
    q.h(0)
    for i in range(n - 1):
        q.cnot(i, i + 1)  # GHZ: 2 amplitudes
    assert abs(q.prob(n - 1) - 0.5) < 1e-6
    r = q.m_all()
    assert r in (0, (1 << n) - 1)


def test_sparse_alu():
    q = make_sp(44, seed=3)
    q.x(0)
    q.x(2)
    q.inc(3, 0, 40)
    assert q.m_reg(0, 40) == 8
    q2 = make_sp(16, seed=4)
    q2.x(1)
    q2.x(2)
    q2.mul_mod_n_out(7, 15, 0, 4, 4)
    assert (q2.m_all() >> 4) == 12


def test_sparse_measure_and_collapse():
    q = make_sp(30, seed=5)
    q.h(0)
    q.cnot(0, 29)
    res = q.multi_shot_measure_mask([1, 1 << 29], 300)
    assert sum(res.values()) == 300
    assert set(res.keys()) <= {0, 3}
    r = q.force_m(0, True)
    assert r
    assert abs(q.prob(29) - 1.0) < 1e-6


def test_sparse_compose_dispose():
    a = make_sp(2, seed=1)
    a.h(0)
    b = make_sp(1, seed=2)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 3
    assert abs(a.prob(2) - 1.0) < 1e-6
    a.dispose_perm(2, 1, 1)
    assert a.num_qubits == 2
    assert abs(a.prob(0) - 0.5) < 1e-5


def test_sparse_under_qunit():
    q = qa.create_simulator(30, layers=["qunit", "sparse"], seed=5)
    q.h(0)
    q.cnot(0, 15)
    q.cnot(15, 29)
    assert abs(q.prob(29) - 0.5) < 1e-5


def test_sparse_swap_block_gates():
    """Regression: swap-block Apply2x2 (offset1 and offset2 both single DIFFERENT
    bits) — fsim/sqrt-swap were mispaired in the sparse general path."""
    import numpy as np
    for gate, args in (("sqrt_swap", (1, 3)), ("isqrt_swap", (0, 2)),
                       ("fsim", (1.41, 0.54, 2, 3))):
        q = qa.create_simulator(4, engine="sparse", seed=3)
        cp = qa.create_simulator(4, engine="cpu", seed=3)
        for s in (q, cp):
            for i in range(4):
                s.ry(0.5 + 0.3 * i, i)
            getattr(s, gate)(*args)
        fid = abs(np.vdot(np.asarray(cp.get_state_vector()),
                          np.asarray(q.get_state_vector())))
        assert fid > 1 - 1e-5, (gate, fid)
