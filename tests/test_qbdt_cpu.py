"""QBdt (binary decision tree) tests vs the dense CPU engine.

Parity model: /root/reference/src/qbdt/* behavior (compressed tree state,
same gate semantics as every other backend).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def make_bdt(n, seed=7):
    return qa.create_simulator(n, layers=["bdt"], seed=seed)


def make_cpu(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


def test_basis_and_single_qubit():
    q = make_bdt(4)
    cp = make_cpu(4)
    q.h(0)
    cp.h(0)
    q.x(2)
    cp.x(2)
    q.t(0)
    cp.t(0)
    q.ry(0.8, 3)
    cp.ry(0.8, 3)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_ghz_product_structure_stays_compressed():
    n = 12
    q = make_bdt(n)
    q.h(0)
    for i in range(n - 1):
        q.cnot(i, i + 1)
    # GHZ is two branches: node count stays linear in n
    # (the compression property the representation exists for)
    sv_nodes = None
    cp = make_cpu(n)
    cp.h(0)
    for i in range(n - 1):
        cp.cnot(i, i + 1)
    assert abs(q.prob(n - 1) - 0.5) < 1e-5
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_control_above_target():
    # control deeper (higher index) than target: the PairMix deep-control path
    q = make_bdt(3)
    cp = make_cpu(3)
    q.h(2)
    cp.h(2)
    q.cnot(2, 0)
    cp.cnot(2, 0)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_anti_control():
    q = make_bdt(2)
    cp = make_cpu(2)
    q.h(1)
    cp.h(1)
    q.anti_cnot(1, 0)
    cp.anti_cnot(1, 0)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_ccnot_mixed_depths():
    q = make_bdt(4)
    cp = make_cpu(4)
    q.h(0)
    cp.h(0)
    q.h(3)
    cp.h(3)
    q.ccnot(0, 3, 1)  # one control above, one below the target
    cp.ccnot(0, 3, 1)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


@pytest.mark.parametrize("seed", [1, 2, 3])
def test_random_circuit_vs_dense(seed):
    n = 5
    rng = np.random.default_rng(seed)
    q = make_bdt(n, seed=seed)
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
            a, b = rng.choice(n, 2, replace=False)
            q.cz(int(a), int(b))
            cp.cz(int(a), int(b))
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)


def test_measurement_collapse():
    q = make_bdt(3, seed=5)
    q.h(0)
    q.cnot(0, 1)
    r0 = q.m(0)
    assert abs(q.prob(1) - (1.0 if r0 else 0.0)) < 1e-5


def test_swap_via_gates():
    q = make_bdt(3, seed=2)
    cp = make_cpu(3, seed=2)
    q.x(0)
    cp.x(0)
    q.swap(0, 2)
    cp.swap(0, 2)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_compose():
    a = make_bdt(2, seed=1)
    a.h(0)
    b = make_bdt(1, seed=2)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 3
    assert abs(a.prob(2) - 1.0) < 1e-5
    assert abs(a.prob(0) - 0.5) < 1e-5


def test_amplitude_access():
    q = make_bdt(3, seed=3)
    q.h(0)
    q.cnot(0, 2)
    a0 = q.get_amplitude(0)
    a5 = q.get_amplitude(5)
    assert abs(abs(a0) - 1 / np.sqrt(2)) < 1e-6
    assert abs(abs(a5) - 1 / np.sqrt(2)) < 1e-6
    assert abs(q.get_amplitude(1)) < 1e-9


def test_qft_small_on_bdt():
    n = 5
    q = make_bdt(n, seed=4)
    cp = make_cpu(n, seed=4)
    for i in range(n):
        if i % 2:
            q.x(i)
            cp.x(i)
    q.qft(0, n)
    cp.qft(0, n)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)


def test_mirror_on_bdt():
    n = 4
    rng = np.random.default_rng(9)
    q = make_bdt(n, seed=9)
    ops = []
    for _ in range(15):
        if rng.random() < 0.6:
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            ops.append(("ry", th, t))
        else:
            a, b = rng.choice(n, 2, replace=False)
            q.cnot(int(a), int(b))
            ops.append(("cnot", int(a), int(b)))
    for op in reversed(ops):
        if op[0] == "ry":
            q.ry(-op[1], op[2])
        else:
            q.cnot(op[1], op[2])
    assert q.m_all() == 0


def test_bdt_hybrid_switches_to_engine():
    import os

    os.environ["QRACK_QBDT_HYBRID_THRESHOLD"] = "0.1"
    try:
        n = 6
        rng = np.random.default_rng(3)
        q = qa.create_simulator(n, layers=["bdt_hybrid", "cpu"], seed=3)
        cp = make_cpu(n, seed=3)
        # random dense circuit: tree blows past 0.1 * 2^6 nodes and switches
        for _ in range(15):
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            cp.ry(th, t)
            a, b = rng.choice(n, 2, replace=False)
            q.cnot(int(a), int(b))
            cp.cnot(int(a), int(b))
        assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)
    finally:
        del os.environ["QRACK_QBDT_HYBRID_THRESHOLD"]


def test_qbdt_branch_rounding_degrades_gracefully():
    """with QRACK_QBDT_SEPARABILITY_THRESHOLD set,
    a random circuit that would exceed the exact node budget completes with
    reported fidelity < 1 instead of raising the RAM guard."""
    import os

    import numpy as np

    # weakly-entangling circuit: small rotations + CNOT chains produce many
    # low-weight branches — exactly what rounding absorbs; the exact tree
    # still exceeds the node cap
    os.environ["QRACK_QBDT_MAX_NODES"] = "200"
    os.environ["QRACK_QBDT_SEPARABILITY_THRESHOLD"] = "0.05"
    try:
        n = 12
        rng = np.random.default_rng(7)
        q = qa.create_simulator(n, layers=["bdt"], seed=7)
        for layer in range(8):
            for t in range(n):
                th = float(rng.uniform(0.05, 0.25))
                q.ry(th, t)
            for t in range(0, n - 1):
                q.cnot(t, t + 1)
        # completed under a node cap that the exact tree blows through
        fid = q.get_unitary_fidelity()
        assert 0.0 < fid <= 1.0
        p = q.prob(0)
        assert 0.0 <= p <= 1.0 + 1e-6
    finally:
        del os.environ["QRACK_QBDT_MAX_NODES"]
        del os.environ["QRACK_QBDT_SEPARABILITY_THRESHOLD"]
    # and without rounding, the same circuit must hit the guard
    os.environ["QRACK_QBDT_MAX_NODES"] = "200"
    try:
        q2 = qa.create_simulator(n, layers=["bdt"], seed=7)
        rng = np.random.default_rng(7)
        with pytest.raises(Exception):
            for layer in range(8):
                for t in range(n):
                    th = float(rng.uniform(0.05, 0.25))
                    q2.ry(th, t)
                for t in range(0, n - 1):
                    q2.cnot(t, t + 1)
    finally:
        del os.environ["QRACK_QBDT_MAX_NODES"]


def test_qbdt_rounding_accuracy_small_threshold():
    """Tiny thresholds keep the state numerically close to exact."""
    import os

    import numpy as np

    os.environ["QRACK_QBDT_SEPARABILITY_THRESHOLD"] = "1e-9"
    try:
        n = 6
        q = qa.create_simulator(n, layers=["bdt"], seed=3)
        cp = qa.create_simulator(n, engine="cpu", seed=3)
        rng = np.random.default_rng(3)
        for _ in range(30):
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            cp.ry(th, t)
            c = int(rng.integers(n))
            if c != t:
                q.cz(c, t)
                cp.cz(c, t)
        sv1 = np.asarray(q.get_state_vector())
        sv2 = np.asarray(cp.get_state_vector())
        f = abs(np.vdot(sv1, sv2))
        assert f > 1.0 - 1e-5
    finally:
        del os.environ["QRACK_QBDT_SEPARABILITY_THRESHOLD"]
