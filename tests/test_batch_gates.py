"""Batched independent single-qubit gates (Mtrx1qBatch fused pass)."""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close

def _rand_u2(rng):
    # Haar-ish random 2x2 unitary
    th, ph, lm = rng.uniform(0, 2 * np.pi, 3)
    c, s = np.cos(th / 2), np.sin(th / 2)
    return [c, -np.exp(1j * lm) * s, np.exp(1j * ph) * s, np.exp(1j * (ph + lm)) * c]


@pytest.mark.parametrize("k", [2, 3, 5, 7])
@pytest.mark.parametrize("precision", ["fp32", "fp64"])
def test_mtrx_1q_batch_matches_sequential(k, precision):
    n = 8
    rng = np.random.default_rng(100 + k)
    targets = list(rng.choice(n, k, replace=False).astype(int))
    ms = [_rand_u2(rng) for _ in range(k)]
    qb = qa.create_simulator(n, engine="cpu", precision=precision, seed=1)
    qs = qa.create_simulator(n, engine="cpu", precision=precision, seed=1)
    for i in range(n):
        qb.ry(0.3 + 0.1 * i, i)
        qs.ry(0.3 + 0.1 * i, i)
    flat = [complex(x) for m in ms for x in m]
    qb.mtrx_1q_batch(targets, flat)
    for t, m in zip(targets, ms):
        qs.mtrx([complex(x) for x in m], t)
    assert_states_close(qb.get_state_vector(), qs.get_state_vector(), 1e-4)


def test_mtrx_1q_batch_duplicate_targets_sequential_order():
    # duplicates fall back to in-order sequential application
    n = 3
    q1 = qa.create_simulator(n, engine="cpu", seed=1)
    q2 = qa.create_simulator(n, engine="cpu", seed=1)
    h = [0.7071067811865476] * 3 + [-0.7071067811865476]
    s_gate = [1, 0, 0, 1j]
    q1.mtrx_1q_batch([0, 0], [complex(x) for x in h + s_gate])
    q2.h(0)
    q2.s(0)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-6)


def test_mtrx_1q_batch_on_layered_stack():
    # layered sims lower the batch per-gate; result must match
    n = 5
    rng = np.random.default_rng(7)
    targets = [0, 2, 4]
    ms = [_rand_u2(rng) for _ in targets]
    flat = [complex(x) for m in ms for x in m]
    qu = qa.create_simulator(n, layers=["qunit", "cpu"], seed=2)
    cp = qa.create_simulator(n, engine="cpu", seed=2)
    qu.mtrx_1q_batch(targets, flat)
    cp.mtrx_1q_batch(targets, flat)
    assert_states_close(qu.get_state_vector(), cp.get_state_vector(), 1e-5)


@pytest.mark.parametrize("layers", [
    ["qunit", "cpu"],
    ["qunit", "stabilizer_hybrid", "cpu"],
    ["pager", "cpu"],
    ["stabilizer_hybrid", "cpu"],
])
def test_batch_routes_through_stacks(layers):
    # entangle first so QUnit holds merged units, then batch a full 1q layer
    n = 6
    rng = np.random.default_rng(21)
    qb = qa.create_simulator(n, layers=layers, seed=4, pages_per_device=2)
    qs = qa.create_simulator(n, engine="cpu", seed=4)
    for s in (qb, qs):
        for i in range(n):
            s.ry(0.4 + 0.1 * i, i)
        for i in range(0, n - 1, 2):
            s.cnot(i, i + 1)
    targets = list(range(n))
    ms = [_rand_u2(rng) for _ in targets]
    flat = [complex(x) for m in ms for x in m]
    qb.mtrx_1q_batch(targets, flat)
    for t, m in zip(targets, ms):
        qs.mtrx([complex(x) for x in m], t)
    assert_states_close(qb.get_state_vector(), qs.get_state_vector(), 1e-4)


@pytest.mark.parametrize("precision", ["fp32", "fp64"])
def test_cnot_batch_matches_sequential(precision):
    n = 8
    rng = np.random.default_rng(31)
    for trial in range(3):
        perm = rng.permutation(n)
        controls = [int(perm[i]) for i in range(0, n, 2)]
        targets = [int(perm[i + 1]) for i in range(0, n, 2)]
        qb = qa.create_simulator(n, engine="cpu", precision=precision, seed=5)
        qs = qa.create_simulator(n, engine="cpu", precision=precision, seed=5)
        for i in range(n):
            th = float(rng.uniform(0, np.pi))
            qb.ry(th, i)
            qs.ry(th, i)
        qb.cnot_batch(controls, targets)
        for c, t in zip(controls, targets):
            qs.cnot(c, t)
        assert_states_close(qb.get_state_vector(), qs.get_state_vector(), 1e-5)


def test_cnot_batch_overlapping_falls_back():
    # shared qubit across pairs: sequential semantics (in order)
    q1 = qa.create_simulator(3, engine="cpu", seed=1)
    q2 = qa.create_simulator(3, engine="cpu", seed=1)
    q1.x(0)
    q2.x(0)
    q1.cnot_batch([0, 1], [1, 2])  # chain: overlapping qubit 1
    q2.cnot(0, 1)
    q2.cnot(1, 2)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-6)


@pytest.mark.parametrize("precision", ["fp32", "fp64"])
def test_cphase_pairs_matches_sequential(precision):
    n = 7
    rng = np.random.default_rng(51)
    controls = [0, 2, 4, 1]  # overlapping qubit 1 allowed: diagonal commutes
    targets = [1, 3, 5, 6]
    angles = [float(a) for a in rng.uniform(0.2, 2.8, 4)]
    qb = qa.create_simulator(n, engine="cpu", precision=precision, seed=5)
    qs = qa.create_simulator(n, engine="cpu", precision=precision, seed=5)
    for i in range(n):
        qb.h(i)
        qs.h(i)
    qb.cphase_pairs(controls, targets, angles)
    for c, t, a in zip(controls, targets, angles):
        qs.mcphase([c], 1, complex(np.exp(1j * a)), t)
    assert_states_close(qb.get_state_vector(), qs.get_state_vector(), 1e-5)


def test_cz_batch_graph_state():
    # 1D cluster state: H all + CZ chain in two one-pass layers
    n = 6
    q1 = qa.create_simulator(n, engine="cpu", seed=2)
    q2 = qa.create_simulator(n, engine="cpu", seed=2)
    for i in range(n):
        q1.h(i)
        q2.h(i)
    q1.cz_batch(list(range(0, n - 1, 2)), list(range(1, n, 2)))
    q1.cz_batch(list(range(1, n - 1, 2)), list(range(2, n, 2)))
    for i in range(n - 1):
        q2.cz(i, i + 1)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-5)


def test_fsim_batch_cpu_default():
    # the default lowering applies gates sequentially: equality is exact
    n = 6
    q1 = qa.create_simulator(n, engine="cpu", seed=8)
    q2 = qa.create_simulator(n, engine="cpu", seed=8)
    for i in range(n):
        q1.h(i)
        q2.h(i)
    thetas, phis = [0.4, 1.2, 0.9], [0.7, 0.2, 1.5]
    a, b = [0, 2, 4], [1, 3, 5]
    q1.fsim_batch(thetas, phis, a, b)
    for th, ph, x, y in zip(thetas, phis, a, b):
        q2.fsim(th, ph, x, y)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-5)


def _rand_u4(rng):
    # Haar-ish random 4x4 unitary via QR
    z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
    qm, r = np.linalg.qr(z)
    return qm * (np.diag(r) / np.abs(np.diag(r)))


@pytest.mark.parametrize("layers", [["cpu"], ["qunit", "cpu"], ["stabilizer_hybrid", "cpu"]])
@pytest.mark.parametrize("pair", [(1, 3), (3, 1), (0, 2)])
def test_mtrx_2q_vs_numpy(layers, pair):
    n = 4
    rng = np.random.default_rng(61)
    u = _rand_u4(rng)
    q = qa.create_simulator(n, layers=layers, precision="fp64", seed=5)
    cp = qa.create_simulator(n, engine="cpu", precision="fp64", seed=5)
    for s in (q, cp):
        for i in range(n):
            s.ry(0.3 + 0.2 * i, i)
    q.mtrx_2q([complex(x) for x in u.flatten()], pair[0], pair[1])
    # numpy reference: basis |q2 q1> on (pair[0]=q1, pair[1]=q2)
    sv = np.asarray(cp.get_state_vector()).astype(np.complex128)
    q1b, q2b = pair
    out = sv.copy()
    for i0 in range(1 << n):
        if (i0 >> q1b) & 1 or (i0 >> q2b) & 1:
            continue
        idx = [i0, i0 | (1 << q1b), i0 | (1 << q2b), i0 | (1 << q1b) | (1 << q2b)]
        vec = sv[idx]
        out[idx] = u @ vec
    got = np.asarray(q.get_state_vector()).astype(np.complex128)
    assert np.abs(got - out).max() < 1e-9


def test_mtrx_2q_batch_cpu_default():
    n = 6
    rng = np.random.default_rng(81)
    us = [_rand_u4(rng) for _ in range(3)]
    q1s, q2s = [0, 2, 5], [1, 3, 4]
    flat = [complex(x) for u in us for x in u.flatten()]
    qb = qa.create_simulator(n, engine="cpu", precision="fp64", seed=6)
    qs = qa.create_simulator(n, engine="cpu", precision="fp64", seed=6)
    for s in (qb, qs):
        for i in range(n):
            s.ry(0.2 + 0.1 * i, i)
    qb.mtrx_2q_batch(flat, q1s, q2s)
    for u, a, b in zip(us, q1s, q2s):
        qs.mtrx_2q([complex(x) for x in u.flatten()], a, b)
    assert_states_close(qb.get_state_vector(), qs.get_state_vector(), 1e-9)
