"""Per-gate correctness vs the numpy reference simulator (CPU engine).

Parity model: /root/reference/test/tests.cpp gate unit tests
(test_apply_single_bit, test_fsim, controlled-gate family).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import RefSim, assert_states_close


def rand_unitary_2x2(rng):
    # Haar-ish: QR of a random complex matrix
    m = rng.normal(size=(2, 2)) + 1j * rng.normal(size=(2, 2))
    q, r = np.linalg.qr(m)
    q = q * (np.diag(r) / np.abs(np.diag(r)))
    return q


@pytest.fixture(params=["fp32", "fp64"])
def precision(request):
    return request.param


def make(n, precision, seed=7):
    return qa.create_simulator(n, precision=precision, engine="cpu", seed=seed)


def prep_random(q, ref, rng, n):
    """Apply a randomizing layer so tests don't run on |0...0>."""
    for i in range(n):
        th = rng.uniform(0, 2 * np.pi)
        q.ry(th, i)
        ref.ry(th, i)
    for i in range(n - 1):
        q.cnot(i, i + 1)
        ref.x(i + 1, controls=[i])


def test_named_single_qubit_gates(precision):
    n = 4
    rng = np.random.default_rng(11)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    q.x(0); ref.x(0)
    q.y(1); ref.y(1)
    q.z(2); ref.z(2)
    q.h(3); ref.h(3)
    q.s(0); ref.s(0)
    q.t(1); ref.t_(1)
    q.rx(0.3, 2); ref.rx(0.3, 2)
    q.ry(1.1, 3); ref.ry(1.1, 3)
    q.rz(2.2, 0); ref.rz(2.2, 0)
    atol = 1e-5 if precision == "fp32" else 1e-10
    assert_states_close(q.get_state_vector(), ref.state, atol)


def test_random_mtrx_gates(precision):
    n = 5
    rng = np.random.default_rng(23)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    for _ in range(20):
        u = rand_unitary_2x2(rng)
        t = int(rng.integers(n))
        q.mtrx(list(u.flatten()), t)
        ref.mtrx(u, t)
    atol = 1e-4 if precision == "fp32" else 1e-9
    assert_states_close(q.get_state_vector(), ref.state, atol)


def test_controlled_gates(precision):
    n = 5
    rng = np.random.default_rng(31)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    q.cnot(0, 1); ref.x(1, controls=[0])
    q.cz(1, 2); ref.z(2, controls=[1])
    q.cy(2, 3); ref.y(3, controls=[2])
    q.ccnot(0, 1, 4); ref.x(4, controls=[0, 1])
    q.anti_cnot(3, 0); ref.x(0, anti=[3])
    u = rand_unitary_2x2(rng)
    q.mcmtrx([2, 4], list(u.flatten()), 0); ref.mtrx(u, 0, controls=[2, 4])
    u2 = rand_unitary_2x2(rng)
    q.macmtrx([1, 3], list(u2.flatten()), 2); ref.mtrx(u2, 2, anti=[1, 3])
    atol = 1e-4 if precision == "fp32" else 1e-9
    assert_states_close(q.get_state_vector(), ref.state, atol)


def test_ucmtrx_mixed_polarity(precision):
    n = 4
    rng = np.random.default_rng(41)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    u = rand_unitary_2x2(rng)
    # controls [0,2], perm=0b01: control 0 must be 1, control 2 must be 0
    q.ucmtrx([0, 2], list(u.flatten()), 3, 0b01)
    ref.mtrx(u, 3, controls=[0], anti=[2])
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_multiplexer(precision):
    n = 4
    rng = np.random.default_rng(43)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    mtrxs = [rand_unitary_2x2(rng) for _ in range(4)]
    flat = np.concatenate([m.flatten() for m in mtrxs]).astype(np.complex128)
    # controls [1,3]: selector bit0=q1, bit1=q3
    q.uniformly_controlled_single_bit([1, 3], 0, flat)
    ref.mtrx(mtrxs[0], 0, anti=[1, 3])
    ref.mtrx(mtrxs[1], 0, controls=[1], anti=[3])
    ref.mtrx(mtrxs[2], 0, controls=[3], anti=[1])
    ref.mtrx(mtrxs[3], 0, controls=[1, 3])
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_swap_family(precision):
    n = 4
    rng = np.random.default_rng(53)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    q.swap(0, 2)
    ref.swap(0, 2)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)
    # sqrt_swap twice == swap
    q.sqrt_swap(1, 3)
    q.sqrt_swap(1, 3)
    ref.swap(1, 3)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)
    # iswap then iiswap == identity
    q.iswap(0, 1)
    q.iiswap(0, 1)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_fsim_against_matrix(precision):
    n = 2
    th, ph = 0.37, 1.21
    q = make(n, precision)
    rng = np.random.default_rng(61)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    q.fsim(th, ph, 0, 1)
    # apply the 4x4 fsim matrix to ref (basis order |q1 q0>)
    m = np.eye(4, dtype=np.complex128)
    c, s = np.cos(th), np.sin(th)
    m[1, 1] = c; m[1, 2] = -1j * s
    m[2, 1] = -1j * s; m[2, 2] = c
    m[3, 3] = np.exp(-1j * ph)
    ref.state = m @ ref.state
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_mask_gates(precision):
    n = 5
    rng = np.random.default_rng(71)
    q = make(n, precision)
    ref = RefSim(n)
    prep_random(q, ref, rng, n)
    q.x_mask(0b10101)
    for t in (0, 2, 4):
        ref.x(t)
    q.z_mask(0b01010)
    for t in (1, 3):
        ref.z(t)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)
    # phase_parity: e^{-i r/2} on even parity, e^{+i r/2} on odd
    r = 0.77
    q.phase_parity(r, 0b111)
    idx = np.arange(1 << n)
    par = ((idx & 1) ^ ((idx >> 1) & 1) ^ ((idx >> 2) & 1)).astype(bool)
    ref.state[par] *= np.exp(1j * r / 2)
    ref.state[~par] *= np.exp(-1j * r / 2)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_global_phase_tracked(precision):
    # Phase gates applied to |0> keep exact global phase in the state vector
    q = make(1, precision)
    q.phase(-1 + 0j, 1 + 0j, 0)
    sv = q.get_state_vector()
    assert abs(sv[0] + 1) < 1e-6


def test_rol_ror(precision):
    n = 4
    q = make(n, precision, seed=3)
    q.x(0)  # |0001>
    q.rol(1, 0, 4)
    assert q.m_all() == 0b0010
    q2 = make(n, precision, seed=3)
    q2.x(0)
    q2.ror(1, 0, 4)
    assert q2.m_all() == 0b1000


def test_time_evolve_pauli_x(precision):
    # exp(-i t X) on |0> = cos t |0> - i sin t |1>
    q = make(1, precision)
    t = 0.6
    q.time_evolve([{"target": 0, "matrix": [0, 1, 1, 0]}], t)
    sv = q.get_state_vector()
    assert abs(sv[0] - np.cos(t)) < 1e-5
    assert abs(sv[1] + 1j * np.sin(t)) < 1e-5


def test_time_evolve_vs_scipy_random_hermitian():
    """TimeEvolve against scipy expm for random Hermitian 2x2 terms,
    including controlled terms."""
    import scipy.linalg as sla

    rng = np.random.default_rng(17)
    for trial in range(6):
        a, b = rng.normal(size=2)
        c = rng.normal() + 1j * rng.normal()
        H = np.array([[a, c], [np.conj(c), b]])
        t = float(rng.uniform(0.1, 1.5))
        U = sla.expm(-1j * H * t)
        q = qa.create_simulator(2, engine="cpu", precision="fp64", seed=1)
        q.ry(0.9, 0)
        q.ry(0.4, 1)
        sv0 = np.asarray(q.get_state_vector()).astype(complex).reshape(2, 2)
        q.time_evolve([{"target": 0, "matrix": [complex(H[0, 0]), complex(H[0, 1]),
                                                complex(H[1, 0]), complex(H[1, 1])]}], t)
        got = np.asarray(q.get_state_vector()).astype(complex).reshape(2, 2)
        # qubit 0 is the LSB: state[i1][i0]; apply U on axis 1
        want = np.einsum("ab,ib->ia", U, sv0)
        assert np.abs(got - want).max() < 1e-6, trial
