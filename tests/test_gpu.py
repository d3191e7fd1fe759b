"""HIP engine (MI355X) tests — run on a GPU box via gpurun.

Each numerics test compares the HIP engine against the plain numpy fp64
reference simulator (ref_sim.RefSim) and/or the CPU engine with identical
seeds. Parity model: the engine-matrix rerun strategy of
/root/reference/test/test_main.cpp (same cases, every backend).
"""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import RefSim, assert_states_close

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(qa.hip_device_count() == 0, reason="no HIP device")


def make(n, seed=7, precision="fp32"):
    assert qa.hip_device_count() > 0, "HIP engine must be present on GPU box (no silent fallback)"
    return qa.create_simulator(n, precision=precision, engine="hip", seed=seed)


def rand_unitary_2x2(rng):
    m = rng.normal(size=(2, 2)) + 1j * rng.normal(size=(2, 2))
    q, r = np.linalg.qr(m)
    return q * (np.diag(r) / np.abs(np.diag(r)))


@pytest.mark.parametrize("precision", ["fp32", "fp64"])
def test_gate_numerics_vs_reference(precision):
    n = 10
    rng = np.random.default_rng(101)
    q = make(n, precision=precision)
    ref = RefSim(n)
    for layer in range(8):
        for i in range(n):
            u = rand_unitary_2x2(rng)
            q.mtrx(list(u.flatten()), i)
            ref.mtrx(u, i)
        for i in range(0, n - 1, 2):
            q.cnot(i, i + 1)
            ref.x(i + 1, controls=[i])
        c, t = rng.choice(n, 2, replace=False)
        q.cz(int(c), int(t))
        ref.z(int(t), controls=[int(c)])
    atol = 2e-4 if precision == "fp32" else 1e-9
    assert_states_close(q.get_state_vector(), ref.state, atol)


def test_phase_invert_fast_paths():
    n = 8
    rng = np.random.default_rng(103)
    q = make(n)
    ref = RefSim(n)
    for i in range(n):
        q.h(i)
        ref.h(i)
    for _ in range(20):
        t = int(rng.integers(n))
        which = rng.integers(4)
        if which == 0:
            q.z(t); ref.z(t)
        elif which == 1:
            q.s(t); ref.s(t)
        elif which == 2:
            q.x(t); ref.x(t)
        else:
            q.y(t); ref.y(t)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_controlled_and_swap():
    n = 8
    rng = np.random.default_rng(107)
    q = make(n)
    ref = RefSim(n)
    for i in range(n):
        th = rng.uniform(0, np.pi)
        q.ry(th, i)
        ref.ry(th, i)
    q.ccnot(0, 1, 2); ref.x(2, controls=[0, 1])
    q.swap(3, 6); ref.swap(3, 6)
    q.mcmtrx([4, 5], list(rand_unitary_2x2(np.random.default_rng(1)).flatten()), 7)
    ref.mtrx(rand_unitary_2x2(np.random.default_rng(1)), 7, controls=[4, 5])
    q.sqrt_swap(0, 7)
    q.sqrt_swap(0, 7)
    ref.swap(0, 7)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_multiplexer_gpu():
    n = 6
    rng = np.random.default_rng(109)
    q = make(n)
    ref = RefSim(n)
    for i in range(n):
        q.h(i)
        ref.h(i)
    mtrxs = [rand_unitary_2x2(rng) for _ in range(4)]
    flat = np.concatenate([m.flatten() for m in mtrxs]).astype(np.complex128)
    q.uniformly_controlled_single_bit([1, 3], 0, flat)
    ref.mtrx(mtrxs[0], 0, anti=[1, 3])
    ref.mtrx(mtrxs[1], 0, controls=[1], anti=[3])
    ref.mtrx(mtrxs[2], 0, controls=[3], anti=[1])
    ref.mtrx(mtrxs[3], 0, controls=[1, 3])
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_mask_gates_gpu():
    n = 8
    q = make(n)
    ref = RefSim(n)
    for i in range(n):
        q.h(i)
        ref.h(i)
    q.x_mask(0b1010101)
    for t in (0, 2, 4, 6):
        ref.x(t)
    q.z_mask(0b0110)
    ref.z(1)
    ref.z(2)
    q.phase_parity(0.9, 0b11011)
    idx = np.arange(1 << n)
    par = np.zeros(1 << n, dtype=bool)
    for b in (0, 1, 3, 4):
        par ^= ((idx >> b) & 1).astype(bool)
    ref.state[par] *= np.exp(1j * 0.45)
    ref.state[~par] *= np.exp(-1j * 0.45)
    assert_states_close(q.get_state_vector(), ref.state, 1e-4)


def test_prob_and_measure_gpu():
    q = make(12, seed=5)
    q.h(0)
    q.cnot(0, 1)
    assert abs(q.prob(1) - 0.5) < 1e-5
    assert abs(q.prob_mask(0b11, 0b11) - 0.5) < 1e-5
    assert abs(q.prob_parity(0b11)) < 1e-5
    r = q.force_m(0, True)
    assert r
    assert abs(q.prob(1) - 1.0) < 1e-5


def test_m_all_and_multishot_gpu():
    q = make(10, seed=6)
    q.h(0)
    q.cnot(0, 1)
    res = q.multi_shot_measure_mask([1, 2], 1000)
    assert sum(res.values()) == 1000
    assert set(res.keys()) <= {0, 3}
    assert 350 < res.get(0, 0) < 650
    r = q.m_all()
    assert r in (0, 3)


def test_qft_roundtrip_gpu():
    n = 12
    rng = np.random.default_rng(11)
    q = make(n, seed=11)
    for i in range(n):
        q.ry(float(rng.uniform(0, np.pi)), i)
    for i in range(n - 1):
        q.cnot(i, i + 1)
    before = q.get_state_vector()
    q.qft(0, n)
    q.iqft(0, n)
    after = q.get_state_vector()
    assert np.allclose(before, after, atol=1e-3)


def test_alu_gpu():
    q = make(8)
    for i in (0, 2):  # reg = 5
        q.x(i)
    q.inc(3, 0, 4)
    assert q.m_reg(0, 4) == 8
    q2 = make(8)
    q2.x(0)
    q2.x(1)  # 3
    q2.mul(5, 0, 4, 4)
    assert q2.m_reg(0, 8) == 15
    q3 = make(8)
    q3.x(1)
    q3.x(2)  # in = 6
    q3.mul_mod_n_out(7, 15, 0, 4, 4)
    r = q3.m_all()
    assert (r >> 4) == 12
    q4 = make(8)
    q4.x(0)
    q4.x(1)  # 3
    q4.pow_mod_n_out(2, 15, 0, 4, 4)
    assert (q4.m_all() >> 4) == 8


def test_alu_superposition_gpu():
    q = make(6, seed=9)
    q.h(0)
    q.inc(1, 0, 3)
    sv = q.get_state_vector()
    assert abs(abs(sv[1]) - 1 / np.sqrt(2)) < 1e-4
    assert abs(abs(sv[2]) - 1 / np.sqrt(2)) < 1e-4


def test_indexed_lda_gpu():
    table = bytes([10, 20, 30, 40])
    q = make(8)
    q.x(1)  # index 2
    q.indexed_lda(0, 2, 2, 6, table)
    assert (q.m_all() >> 2) == 30


def test_hash_gpu():
    table = bytes([2, 0, 3, 1])
    q = make(2, seed=3)
    q.h(0)
    q.hash(0, 2, table)
    sv = q.get_state_vector()
    assert abs(abs(sv[2]) - 1 / np.sqrt(2)) < 1e-4
    assert abs(abs(sv[0]) - 1 / np.sqrt(2)) < 1e-4


def test_compose_decompose_gpu():
    a = make(2, seed=1)
    a.h(0)
    b = make(1, seed=2)
    b.x(0)
    a.compose(b)
    assert a.num_qubits == 3
    assert abs(a.prob(2) - 1.0) < 1e-5
    dest = make(1, seed=3)
    a.decompose(2, dest)
    assert a.num_qubits == 2
    assert abs(dest.prob(0) - 1.0) < 1e-5
    assert abs(a.prob(0) - 0.5) < 1e-5


def test_decompose_entangled_part_gpu():
    q = make(4, seed=5)
    q.h(1)
    q.cnot(1, 2)
    q.h(0)
    dest = make(2, seed=6)
    q.decompose(1, dest)
    assert abs(dest.prob_mask(0b11, 0b00) - 0.5) < 1e-4
    assert abs(dest.prob_mask(0b11, 0b11) - 0.5) < 1e-4
    assert abs(q.prob(0) - 0.5) < 1e-4


def test_dispose_gpu():
    q = make(4, seed=7)
    q.h(0)
    q.x(2)
    q.dispose(2, 1)
    assert q.num_qubits == 3
    assert abs(q.prob(0) - 0.5) < 1e-4


def test_allocate_clone_gpu():
    q = make(3, seed=8)
    q.h(0)
    q.allocate(2)
    assert q.num_qubits == 5
    c = q.clone()
    c.x(4)
    assert abs(q.prob(4)) < 1e-5
    assert abs(c.prob(4) - 1.0) < 1e-5
    assert not q.approx_compare(c)


def test_cpu_gpu_equivalence():
    """Same circuit, same seed -> same measurement outcomes and states."""
    n = 10
    rng = np.random.default_rng(77)
    qc = qa.create_simulator(n, engine="cpu", seed=55)
    qg = make(n, seed=55)
    for layer in range(5):
        for i in range(n):
            th = float(rng.uniform(0, 2 * np.pi))
            qc.ry(th, i)
            qg.ry(th, i)
        for i in range(n - 1):
            qc.cnot(i, i + 1)
            qg.cnot(i, i + 1)
    svc = np.asarray(qc.get_state_vector())
    svg = np.asarray(qg.get_state_vector())
    assert np.allclose(svc, svg, atol=2e-4)
    assert qc.m_all() == qg.m_all()


def test_expectation_gpu():
    q = make(5, seed=3)
    q.x(1)
    q.h(0)
    assert abs(q.expectation_bits_all([0, 1, 2]) - 2.5) < 1e-4
    assert abs(q.variance_bits_all([0, 1, 2]) - 0.25) < 1e-4


def test_phase_flip_if_less_gpu():
    q = make(3, seed=4)
    q.h(0)
    q.h(1)
    q.phase_flip_if_less(2, 0, 3)
    sv = q.get_state_vector()
    assert sv[0].real < 0 and sv[1].real < 0
    assert sv[2].real > 0


def test_sum_sqr_diff_gpu():
    a = make(6, seed=1)
    b = make(6, seed=2)
    a.h(0)
    b.h(0)
    assert a.sum_sqr_diff(b) < 1e-5
    b.x(3)
    assert a.sum_sqr_diff(b) > 1.0


def test_native_extension_loaded():
    """Fail loudly if the native HIP path is absent on a GPU box."""
    import qrack_amd._qrack as native

    assert native.hip_device_count() > 0
    assert "_qrack" in native.__file__


def test_mirror_circuit_24q_gpu():
    """Deep mirror circuit at 24 qubits on the HIP engine: random layer +
    inverse returns to the initial basis state (the reference's [mirror]
    self-validation strategy at GPU scale)."""
    n = 24
    rng = np.random.default_rng(77)
    q = make(n, seed=77)
    init = int(rng.integers(1 << n))
    for i in range(n):
        if (init >> i) & 1:
            q.x(i)
    ops = []
    for _ in range(60):
        kind = rng.integers(4)
        if kind == 0:
            t = int(rng.integers(n))
            th = float(rng.uniform(0, 2 * np.pi))
            q.ry(th, t)
            ops.append(("ry", th, t))
        elif kind == 1:
            a, b = rng.choice(n, 2, replace=False)
            q.cnot(int(a), int(b))
            ops.append(("cnot", int(a), int(b)))
        elif kind == 2:
            t = int(rng.integers(n))
            q.t(t)
            ops.append(("t", t))
        else:
            a, b = rng.choice(n, 2, replace=False)
            q.swap(int(a), int(b))
            ops.append(("swap", int(a), int(b)))
    for op in reversed(ops):
        if op[0] == "ry":
            q.ry(-op[1], op[2])
        elif op[0] == "cnot":
            q.cnot(op[1], op[2])
        elif op[0] == "t":
            q.it(op[1])
        else:
            q.swap(op[1], op[2])
    assert q.m_all() == init


def test_large_width_30q_invariants_gpu():
    """Large-width regression guard: 30 qubits is
    past the 2^31 work-item wrap where the exact-grid/32-bit-index bug class
    lives (fixed in commit a94602b). Norm stays 1 through a mixing layer and
    a mirror sequence returns to the initial permutation, asserted in-suite
    rather than by probe logs."""
    n = 30
    q = make(n, seed=31)
    init = 0x2A5A5A5A & ((1 << n) - 1)
    q.set_permutation(init)
    ops = []
    rng = np.random.default_rng(31)
    # touch low, mid, AND top qubits so index arithmetic at 2^29+ pairs runs
    for t in [0, 1, 14, 15, 27, 28, 29]:
        q.h(t)
        ops.append(("h", t))
        th = float(rng.uniform(0, 2 * np.pi))
        q.rz(th, t)
        ops.append(("rz", th, t))
    for a, b in [(0, 29), (14, 28), (1, 27)]:
        q.cnot(a, b)
        ops.append(("cnot", a, b))
    # probabilities well-defined (norm ~ 1): sum over one qubit's marginal
    p0 = q.prob(29)
    assert 0.0 <= p0 <= 1.0 + 1e-5
    for op in reversed(ops):
        if op[0] == "h":
            q.h(op[1])
        elif op[0] == "rz":
            q.rz(-op[1], op[2])
        else:
            q.cnot(op[1], op[2])
    assert q.m_all() == init


@pytest.mark.parametrize("precision", ["fp32", "fp64"])
def test_mtrx_1q_batch_gpu(precision):
    # fused k-gate pass vs sequential application (vector path: no target 0;
    # scalar path: target 0 included; chunking: k=7 > max batch)
    n = 20
    rng = np.random.default_rng(11)
    for targets in ([3, 7, 12, 15, 18], [0, 1, 5, 9], list(range(7))):
        ms = []
        for _ in targets:
            th, ph, lm = rng.uniform(0, 2 * np.pi, 3)
            c, s = np.cos(th / 2), np.sin(th / 2)
            ms.append([c, -s * np.exp(1j * lm), s * np.exp(1j * ph),
                       c * np.exp(1j * (ph + lm))])
        qb = qa.create_simulator(n, engine="hip", precision=precision, seed=3)
        qs = qa.create_simulator(n, engine="hip", precision=precision, seed=3)
        for i in range(0, n, 3):
            qb.h(i)
            qs.h(i)
        qb.mtrx_1q_batch(targets, [complex(x) for m in ms for x in m])
        for t, m in zip(targets, ms):
            qs.mtrx([complex(x) for x in m], t)
        assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_cnot_batch_gpu():
    n = 22
    rng = np.random.default_rng(41)
    for controls, targets in (([2, 8, 14], [5, 11, 19]), ([0, 3], [1, 4])):
        qb = qa.create_simulator(n, engine="hip", seed=6)
        qs = qa.create_simulator(n, engine="hip", seed=6)
        for i in range(0, n, 2):
            qb.h(i)
            qs.h(i)
            qb.t(i)
            qs.t(i)
        qb.cnot_batch(controls, targets)
        for c, t in zip(controls, targets):
            qs.cnot(c, t)
        assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_cphase_pairs_gpu():
    n = 21
    qb = qa.create_simulator(n, engine="hip", seed=7)
    qs = qa.create_simulator(n, engine="hip", seed=7)
    for i in range(n):
        qb.h(i)
        qs.h(i)
    controls = [0, 5, 10, 15]
    targets = [3, 8, 13, 18]
    angles = [0.3, 1.1, 2.2, 0.7]
    qb.cphase_pairs(controls, targets, angles)
    import numpy as np
    for c, t, a in zip(controls, targets, angles):
        qs.mcphase([c], 1, complex(np.exp(1j * a)), t)
    assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_fsim_batch_gpu():
    n = 20
    qb = qa.create_simulator(n, engine="hip", seed=9)
    qs = qa.create_simulator(n, engine="hip", seed=9)
    for i in range(n):
        qb.h(i)
        qs.h(i)
    # mix of in-tile (low) and high pairs
    thetas = [0.4, 1.2, 0.9, 0.3]
    phis = [0.7, 0.2, 1.5, 2.0]
    a = [0, 4, 8, 14]
    b = [1, 5, 9, 17]
    qb.fsim_batch(thetas, phis, a, b)
    for th, ph, x, y in zip(thetas, phis, a, b):
        qs.fsim(th, ph, x, y)
    assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_mtrx_2q_gpu():
    n = 20
    rng = np.random.default_rng(71)
    z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
    qm, r = np.linalg.qr(z)
    u = qm * (np.diag(r) / np.abs(np.diag(r)))
    flat = [complex(x) for x in u.flatten()]
    for pair in ((3, 15), (15, 3), (0, 1)):
        qh = qa.create_simulator(n, engine="hip", seed=5)
        qc = qa.create_simulator(n, engine="cpu", seed=5)
        for s in (qh, qc):
            for i in range(0, n, 2):
                s.h(i)
        qh.mtrx_2q(flat, pair[0], pair[1])
        qc.mtrx_2q(flat, pair[0], pair[1])
        svh = np.asarray(qh.get_state_vector())
        svc = np.asarray(qc.get_state_vector())
        assert np.abs(svh - svc).max() < 1e-5


def test_mtrx_2q_batch_gpu():
    n = 20
    rng = np.random.default_rng(91)
    def u4():
        z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
        qm, r = np.linalg.qr(z)
        return qm * (np.diag(r) / np.abs(np.diag(r)))
    us = [u4() for _ in range(4)]
    q1s, q2s = [0, 5, 9, 14], [3, 6, 2, 18]  # mix of in-tile / high / swapped
    flat = [complex(x) for u in us for x in u.flatten()]
    qb = qa.create_simulator(n, engine="hip", seed=7)
    qs = qa.create_simulator(n, engine="hip", seed=7)
    for s in (qb, qs):
        for i in range(0, n, 3):
            s.h(i)
    qb.mtrx_2q_batch(flat, q1s, q2s)
    for u, a, b in zip(us, q1s, q2s):
        qs.mtrx_2q([complex(x) for x in u.flatten()], a, b)
    assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_mfma_gate_variant():
    # QRACK_GPU_MFMA=1 routes single-target fp32 gates through the MFMA
    # 4x4-real mat-vec kernel; numerics must match the default path
    import subprocess, sys, os
    code = """
import os, sys
sys.path.insert(0, os.getcwd())
import numpy as np, qrack_amd as qa
q = qa.create_simulator(20, engine="hip", seed=3)
rng = np.random.default_rng(5)
for i in range(20):
    q.ry(float(rng.uniform(0, np.pi)), i)
q.h(0); q.h(13); q.t(5); q.x(9)
sv = np.asarray(q.get_state_vector())
np.save("/tmp/mfma_ab.npy", sv)
print("OK")
"""
    env0 = {**os.environ, "QRACK_GPU_MFMA": "0"}
    env1 = {**os.environ, "QRACK_GPU_MFMA": "1"}
    r0 = subprocess.run([sys.executable, "-c", code], env=env0, capture_output=True, text=True)
    assert "OK" in r0.stdout, r0.stderr
    import numpy as np
    ref = np.load("/tmp/mfma_ab.npy")
    r1 = subprocess.run([sys.executable, "-c", code], env=env1, capture_output=True, text=True)
    assert "OK" in r1.stdout, r1.stderr
    got = np.load("/tmp/mfma_ab.npy")
    assert np.abs(ref - got).max() < 1e-5


def test_qft_fused2_numerics_vs_cpu():
    """The 2-column fused QFT kernel (k_qft_col2) must match the CPU engine
    exactly (both directions, odd and even lengths)."""
    for n in (9, 12):
        for init in (0, 0b101101, (1 << n) - 3):
            q = make(n, seed=n)
            cp = qa.create_simulator(n, engine="cpu", seed=n)
            q.set_permutation(init & ((1 << n) - 1))
            cp.set_permutation(init & ((1 << n) - 1))
            q.qft(0, n)
            cp.qft(0, n)
            assert_states_close(q.get_state_vector(), cp.get_state_vector(), 2e-4)
            q.iqft(0, n)
            cp.iqft(0, n)
            assert_states_close(q.get_state_vector(), cp.get_state_vector(), 2e-4)
    # offset register QFT (rampStart > 0)
    q = make(10, seed=3)
    cp = qa.create_simulator(10, engine="cpu", seed=3)
    q.set_permutation(0b1011010010)
    cp.set_permutation(0b1011010010)
    q.qft(3, 6)
    cp.qft(3, 6)
    assert_states_close(q.get_state_vector(), cp.get_state_vector(), 2e-4)


def test_pager_qft_identity_prefix_on_hip():
    """QPager over HIP pages: the identity-prefix QFT shortcut (all intra
    columns as one fused per-page engine ladder, csrc/qpager.cpp QFT/IQFT)
    must match the plain HIP engine, forward and inverse."""
    n = 16
    p = qa.create_simulator(n, layers=["pager", "hip"], seed=9, pages_per_device=4)
    r = make(n, seed=9)
    p.set_permutation(0x9A3C)
    r.set_permutation(0x9A3C)
    for i in range(0, n, 5):
        p.ry(0.4 + i, i)
        r.ry(0.4 + i, i)
    p.qft(0, n)
    r.qft(0, n)
    assert_states_close(p.get_state_vector(), r.get_state_vector(), 2e-4)
    p.iqft(0, n)
    r.iqft(0, n)
    assert_states_close(p.get_state_vector(), r.get_state_vector(), 2e-4)


def test_qft_mid_lds_numerics_vs_cpu():
    """The 2D-tile mid-column LDS QFT kernel (k_qft_mid_lds: fires for
    start-0 registers wider than the low-ladder tile — n > 12 fp32,
    n > 11 fp64) must match the CPU engine in both directions, including
    partial trailing groups (nCols < 6)."""
    for n, prec in ((14, "fp32"), (16, "fp32"), (19, "fp32"), (14, "fp64"), (17, "fp64")):
        rng = np.random.default_rng(n)
        q = qa.create_simulator(n, engine="hip", precision=prec, seed=n)
        cp = qa.create_simulator(n, engine="cpu", precision=prec, seed=n)
        init = int(rng.integers(0, 1 << n))
        q.set_permutation(init)
        cp.set_permutation(init)
        for i in range(0, n, 3):  # non-basis state: superpose a few qubits
            q.ry(0.7 + 0.1 * i, i)
            cp.ry(0.7 + 0.1 * i, i)
        q.qft(0, n)
        cp.qft(0, n)
        tol = 2e-4 if prec == "fp32" else 1e-9
        assert_states_close(q.get_state_vector(), cp.get_state_vector(), tol)
        q.iqft(0, n)
        cp.iqft(0, n)
        assert_states_close(q.get_state_vector(), cp.get_state_vector(), tol)


def test_fuser_over_hip_numerics():
    """QFuser batching over the HIP engine matches the bare engine."""
    n = 12
    rng = np.random.default_rng(91)
    f = qa.create_simulator(n, layers=["fuser", "hip"], seed=91)
    b = make(n, seed=91)
    for d in range(5):
        for t in range(n):
            u = rand_unitary_2x2(rng)
            f.mtrx(list(u.flatten()), t)
            b.mtrx(list(u.flatten()), t)
        for a in range(d % 2, n - 1, 2):
            f.cnot(a, a + 1)
            b.cnot(a, a + 1)
    assert_states_close(f.get_state_vector(), b.get_state_vector(), 2e-4)


def test_mtrx2q_pair2_high_bits():
    """Two disjoint HIGH-bit 4x4s must route through the 16-amplitude-orbit
    pair kernel and match per-gate application."""
    n = 18
    rng = np.random.default_rng(87)

    def u4():
        z = rng.normal(size=(4, 4)) + 1j * rng.normal(size=(4, 4))
        qm, r = np.linalg.qr(z)
        return qm * (np.diag(r) / np.abs(np.diag(r)))

    for pairs in [[(13, 16), (14, 17)], [(16, 13), (12, 15)],
                  [(2, 14), (15, 5), (13, 16), (17, 3)]]:
        us = [u4() for _ in pairs]
        qb = make(n, seed=3)
        qs = make(n, seed=3)
        for i in range(n):
            th = float(rng.uniform(0, np.pi))
            qb.ry(th, i)
            qs.ry(th, i)
        flat = [complex(x) for u in us for x in u.flatten()]
        qb.mtrx_2q_batch(flat, [a for a, _ in pairs], [b for _, b in pairs])
        for u, (a, b) in zip(us, pairs):
            qs.mtrx_2q([complex(x) for x in u.flatten()], a, b)
        assert float(qb.sum_sqr_diff(qs)) < 1e-5


def test_qhybrid_paged_promotion_on_gpu():
    """QHybrid's third tier: with QRACK_MAX_PAGE_QB forced low, a 16-qubit
    hybrid sim promotes to QPager over HIP pages and must match the plain
    engine (csrc/qhybrid.hpp three-tier migration)."""
    import os as _os
    old = _os.environ.get("QRACK_MAX_PAGE_QB")
    _os.environ["QRACK_MAX_PAGE_QB"] = "14"
    try:
        h = qa.create_simulator(16, layers=["hybrid"], seed=4)
        r = make(16, seed=4)
        h.set_permutation(0x5A5A)
        r.set_permutation(0x5A5A)
        for i in range(0, 16, 3):
            h.ry(0.3 + i, i)
            r.ry(0.3 + i, i)
        h.qft(0, 16)
        r.qft(0, 16)
        assert_states_close(h.get_state_vector(), r.get_state_vector(), 2e-4)
    finally:
        if old is None:
            _os.environ.pop("QRACK_MAX_PAGE_QB", None)
        else:
            _os.environ["QRACK_MAX_PAGE_QB"] = old


def test_variance_bits_all_vs_cpu():
    """Regression for a fuzz-caught intermittent: HIP variance_bits_all
    must match the CPU engine (EXP_PERM args now ride in the kernarg
    segment — no async staging buffers on the query path)."""
    ops = [("t", 3), ("cnot", 2, 4), ("ry", 1.1374093532031537, 3),
           ("ry", 0.24300806879584336, 3), ("ry", 2.280324914256294, 4),
           ("t", 4), ("h", 3), ("ry", 2.439288787613517, 4), ("t", 2),
           ("cnot", 1, 2), ("cnot", 4, 2)]
    q = make(5, seed=3)
    cp = qa.create_simulator(5, engine="cpu", seed=3)
    for op in ops:
        getattr(q, op[0])(*op[1:])
        getattr(cp, op[0])(*op[1:])
    for _ in range(10):
        assert abs(q.variance_bits_all([0, 1, 2]) - cp.variance_bits_all([0, 1, 2])) < 1e-3
        assert abs(q.expectation_bits_all([0, 1, 2, 3, 4])
                   - cp.expectation_bits_all([0, 1, 2, 3, 4])) < 1e-3
