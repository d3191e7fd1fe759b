"""In-process QPager tests over CPU pages (N pages on one host must equal
the single-engine gate semantics — the reference's CI property,
test_main.cpp:277-283).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def make_paged(n, pages=4, seed=7):
    return qa.create_simulator(
        n, layers=["pager", "cpu"], seed=seed, pages_per_device=pages
    )


def make_cpu(n, seed=7):
    return qa.create_simulator(n, engine="cpu", seed=seed)


@pytest.mark.parametrize("pages", [2, 4])
@pytest.mark.parametrize("seed", [1, 2, 3])
def test_random_circuit_matches_single_engine(pages, seed):
    n = 6
    rng = np.random.default_rng(seed)
    q = make_paged(n, pages, seed=seed)
    cp = make_cpu(n, seed=seed)
    for _ in range(25):
        r = rng.random()
        if r < 0.4:
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            cp.ry(th, t)
        elif r < 0.6:
            t = int(rng.integers(n))
            q.t(t)
            cp.t(t)
        elif r < 0.85:
            a, b = rng.choice(n, 2, replace=False)
            q.cnot(int(a), int(b))
            cp.cnot(int(a), int(b))
        else:
            a, b = rng.choice(n, 2, replace=False)
            q.swap(int(a), int(b))
            cp.swap(int(a), int(b))
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-4)


def test_meta_qubit_gates():
    n = 5
    q = make_paged(n, 4)
    cp = make_cpu(n)
    q.h(n - 1)  # top qubit = meta: exchange path
    cp.h(n - 1)
    q.x(n - 2)  # meta: pointer swap
    cp.x(n - 2)
    q.cz(0, n - 1)  # intra-control on meta target phase
    cp.cz(0, n - 1)
    q.cnot(n - 1, 0)  # meta control, intra target
    cp.cnot(n - 1, 0)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_qft_fused_on_pager():
    n = 7
    x = 37
    q = make_paged(n, 4, seed=5)
    q.set_permutation(x)
    q.qft(0, n)
    sv = np.asarray(q.get_state_vector()).astype(np.complex128)
    N = 1 << n
    k = np.arange(N)
    expected = np.exp(2j * np.pi * x * k / N) / np.sqrt(N)
    rev = np.array([int(format(i, f"0{n}b")[::-1], 2) for i in range(N)])
    inner = np.vdot(expected, sv[rev])
    assert abs(abs(inner) - 1.0) < 1e-4


def test_qft_roundtrip_on_pager():
    n = 6
    q = make_paged(n, 4, seed=5)
    q.set_permutation(11)
    q.qft(0, n)
    q.iqft(0, n)
    sv = q.get_state_vector()
    assert abs(abs(sv[11]) - 1.0) < 1e-4


def test_measurement_on_pager():
    n = 5
    q = make_paged(n, 4, seed=6)
    q.h(0)
    q.cnot(0, n - 1)
    assert abs(q.prob(n - 1) - 0.5) < 1e-5
    res = q.multi_shot_measure_mask([1, 1 << (n - 1)], 400)
    assert sum(res.values()) == 400
    assert set(res.keys()) <= {0, 3}
    r = q.m_all()
    assert r in (0, (1 << (n - 1)) | 1)


def test_force_m_meta_on_pager():
    n = 4
    q = make_paged(n, 4, seed=7)
    q.h(n - 1)
    out = q.force_m(n - 1, True)
    assert out is True
    assert abs(q.prob(n - 1) - 1.0) < 1e-5


def test_mask_gates_on_pager():
    n = 5
    q = make_paged(n, 4, seed=8)
    cp = make_cpu(n, seed=8)
    for i in range(n):
        q.h(i)
        cp.h(i)
    q.x_mask(0b11001)
    cp.x_mask(0b11001)
    q.z_mask(0b10110)
    cp.z_mask(0b10110)
    q.phase_parity(0.8, 0b11010)
    cp.phase_parity(0.8, 0b11010)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)


def test_alu_on_pager():
    q = make_paged(6, 2)
    q.x(0)
    q.x(2)  # 5
    q.inc(3, 0, 4)
    assert q.m_reg(0, 4) == 8
    q2 = make_paged(8, 2)
    q2.x(1)
    q2.x(2)  # 6
    q2.mul_mod_n_out(7, 15, 0, 4, 4)
    assert (q2.m_all() >> 4) == 12


def test_compose_on_pager():
    a = make_paged(3, 2, seed=1)
    a.h(0)
    b = make_cpu(1, seed=2)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 4
    assert abs(a.prob(3) - 1.0) < 1e-5
    assert abs(a.prob(0) - 0.5) < 1e-5


def test_swap_meta_meta_on_pager():
    n = 5
    q = make_paged(n, 4, seed=2)
    q.x(n - 1)
    q.swap(n - 1, n - 2)  # pointer relabel
    assert abs(q.prob(n - 2) - 1.0) < 1e-6
    assert abs(q.prob(n - 1)) < 1e-6


def test_anti_meta_controlled_invert():
    # anti-control on one meta qubit, invert target on another meta qubit
    n = 6
    q = make_paged(n, 4, seed=3)
    cp = make_cpu(n, seed=3)
    for i in range(n):
        q.h(i)
        cp.h(i)
    q.macinvert([n - 1], 1, 1, n - 2)  # anti-CNOT(meta, meta)
    cp.macinvert([n - 1], 1, 1, n - 2)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 1e-5)
    q2 = make_paged(n, 4, seed=4)
    cp2 = make_cpu(n, seed=4)
    q2.x(0)
    cp2.x(0)
    q2.macinvert([0], 1, 1, n - 1)  # anti-control intra (off) on meta target: no-op
    cp2.macinvert([0], 1, 1, n - 1)
    assert_states_close(q2.get_state_vector(), cp2.get_state_vector(), 1e-6)


@pytest.mark.parametrize("n,pages", [(8, 2), (10, 4), (12, 8)])
def test_qft_identity_prefix_shortcut_matches_single_engine(n, pages):
    """QPager::QFT's identity-prefix shortcut (all intra columns as one
    fused per-page engine call) must match the single engine for forward,
    inverse, and offset registers (csrc/qpager.cpp QFT/IQFT)."""
    p = make_paged(n, pages, seed=3)
    r = make_cpu(n, seed=3)
    init = 0x2D & ((1 << n) - 1)
    p.set_permutation(init)
    r.set_permutation(init)
    for q in range(0, n, 4):
        p.ry(0.3 + q, q)
        r.ry(0.3 + q, q)
    p.qft(0, n)
    r.qft(0, n)
    assert_states_close(p.get_state_vector(), r.get_state_vector(), 1e-5)
    p.iqft(0, n)
    r.iqft(0, n)
    assert_states_close(p.get_state_vector(), r.get_state_vector(), 1e-5)
    p.qft(2, n - 3)  # offset register: shortcut must NOT fire mid-ladder
    r.qft(2, n - 3)
    assert_states_close(p.get_state_vector(), r.get_state_vector(), 1e-5)
