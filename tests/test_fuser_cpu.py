"""QFuser — transparent gate-fusion decorator (reference QCircuit gate
combining / MpsShard fusion as a standalone layer). Numerics vs the bare
engine on randomized circuits; batching structure asserted indirectly via
exactness through flush boundaries."""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def pair(n, seed):
    f = qa.create_simulator(n, layers=["fuser", "cpu"], seed=seed)
    c = qa.create_simulator(n, engine="cpu", seed=seed)
    return f, c


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
def test_random_circuit_vs_bare(seed):
    n = 6
    rng = np.random.default_rng(40 + seed)
    f, c = pair(n, seed)
    for _ in range(50):
        k = rng.integers(8)
        t = int(rng.integers(n))
        o = int(rng.integers(n))
        if k == 0:
            f.h(t); c.h(t)
        elif k == 1:
            th = float(rng.uniform(0, 2 * np.pi))
            f.ry(th, t); c.ry(th, t)
        elif k == 2:
            f.t(t); c.t(t)
        elif k == 3 and o != t:
            f.cnot(o, t); c.cnot(o, t)
        elif k == 4 and o != t:
            f.cz(o, t); c.cz(o, t)
        elif k == 5 and o != t:
            th = float(rng.uniform(0, 2 * np.pi))
            f.mcphase([o], 1, np.exp(1j * th), t); c.mcphase([o], 1, np.exp(1j * th), t)
        elif k == 6:
            # mid-circuit probability probe forces a flush
            assert abs(f.prob(t) - c.prob(t)) < 1e-6
        else:
            f.x(t); c.x(t)
    assert_states_close(f.get_state_vector(), c.get_state_vector(), 1e-5)


def test_layer_shape_random_benchmark():
    """The benchmark depth-step shape: full 1q layer + disjoint CNOT layer."""
    n = 8
    rng = np.random.default_rng(9)
    f, c = pair(n, 9)
    for d in range(6):
        for t in range(n):
            th = float(rng.uniform(0, 2 * np.pi))
            f.rz(th, t); c.rz(th, t)
            f.h(t); c.h(t)
        for a in range(0, n - 1, 2):
            f.cnot(a, a + 1); c.cnot(a, a + 1)
    assert_states_close(f.get_state_vector(), c.get_state_vector(), 1e-5)


def test_measurement_and_structural_flush():
    f, c = pair(5, 3)
    for s in (f, c):
        s.h(0)
        s.cnot(0, 1)
        s.t(1)
    r1 = f.m(0)
    r2 = c.m(0)
    # same seed -> same draw
    assert r1 == r2
    f.allocate(1)
    assert f.num_qubits == 6
    f.h(5)
    assert 0.0 <= f.prob(5) <= 1.0


def test_mtrx2q_and_swap_through_fuser():
    n = 5
    rng = np.random.default_rng(21)
    z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
    qm, r = np.linalg.qr(z)
    u = qm * (np.diag(r) / np.abs(np.diag(r)))
    f, c = pair(n, 21)
    for s in (f, c):
        s.h(0)
        s.h(2)
        s.mtrx_2q([complex(x) for x in u.flatten()], 1, 3)
        s.swap(0, 4)
        s.cnot(4, 2)
    assert_states_close(f.get_state_vector(), c.get_state_vector(), 1e-5)


def test_fsim_through_fuser():
    import numpy as np

    n = 6
    rng = np.random.default_rng(31)
    f, c = pair(n, 31)
    for layer in range(4):
        for t in range(n):
            th = float(rng.uniform(0, 2 * np.pi))
            f.ry(th, t); c.ry(th, t)
        for a in range(layer % 2, n - 1, 2):
            th, ph = rng.uniform(0, 2 * np.pi, 2)
            f.fsim(float(th), float(ph), a, a + 1)
            c.fsim(float(th), float(ph), a, a + 1)
    assert_states_close(f.get_state_vector(), c.get_state_vector(), 1e-5)
