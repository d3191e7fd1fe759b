"""Expectation / variance / joint-probability family vs numpy reference
(parity: reference qinterface.hpp ExpectationBitsAll..VarianceUnitaryAll
family; tests.cpp test_exp_var)."""

import numpy as np
import pytest

import qrack_amd as qa


def make_state(n=4, seed=3):
    q = qa.create_simulator(n, engine="cpu", precision="fp64", seed=seed)
    rng = np.random.default_rng(seed)
    for i in range(n):
        q.ry(float(rng.uniform(0, np.pi)), i)
    for i in range(n - 1):
        q.cnot(i, i + 1)
    q.t(0)
    q.h(2)
    sv = np.asarray(q.get_state_vector()).astype(np.complex128)
    return q, sv


def joint(sv, bits):
    n = int(np.log2(len(sv)))
    p = np.abs(sv) ** 2
    out = np.zeros(1 << len(bits))
    for i in range(len(sv)):
        idx = 0
        for k, b in enumerate(bits):
            if (i >> b) & 1:
                idx |= 1 << k
        out[idx] += p[i]
    return out


def test_prob_bits_all():
    q, sv = make_state()
    bits = [0, 2, 3]
    got = np.asarray(q.prob_bits_all(bits))
    np.testing.assert_allclose(got, joint(sv, bits), atol=1e-10)


def test_prob_mask_all():
    q, sv = make_state()
    got = np.asarray(q.prob_mask_all(0b1010))
    np.testing.assert_allclose(got, joint(sv, [1, 3]), atol=1e-10)


def test_floats_factorized_expectation_and_variance():
    q, sv = make_state()
    bits = [0, 1, 3]
    w = [0.5, -1.5, 2.0, 0.25, -3.0, 1.0]
    jp = joint(sv, bits)
    # reference semantics: each basis state's value is the PRODUCT of the
    # chosen per-qubit weights (reference qinterface.cpp:771-803)
    vals = np.array([np.prod([w[2 * k + ((p >> k) & 1)] for k in range(3)])
                     for p in range(8)])
    e_ref = float(np.dot(vals, jp))
    v_ref = float(np.dot(vals ** 2, jp) - e_ref ** 2)
    assert q.expectation_floats_factorized(bits, w) == pytest.approx(e_ref, abs=1e-9)
    assert q.variance_floats_factorized(bits, w) == pytest.approx(v_ref, abs=1e-9)


def test_bits_factorized_variance():
    q, sv = make_state()
    bits = [0, 2]
    perms = [3, 5]
    jp = joint(sv, bits)
    vals = np.array([3 * (p & 1) + 5 * ((p >> 1) & 1) for p in range(4)])
    e = np.dot(vals, jp)
    v_ref = float(np.dot(vals ** 2, jp) - e ** 2)
    assert q.variance_bits_factorized(bits, perms) == pytest.approx(v_ref, abs=1e-9)
    # offset does not change variance
    assert q.variance_bits_factorized(bits, perms, 7) == pytest.approx(v_ref, abs=1e-9)


def test_unitary_all_x_basis_matches_pauli():
    q, sv = make_state()
    s2 = 1 / np.sqrt(2)
    h = [s2, s2, s2, -s2]
    for bits in ([1], [0, 3]):
        e_u = q.expectation_unitary_all(bits, h * len(bits))
        # both are the product observable <X x X x ...>
        e_p = q.expectation_pauli_all(bits, [1] * len(bits))  # PauliX
        assert e_u == pytest.approx(e_p, abs=1e-9)
        v_u = q.variance_unitary_all(bits, h * len(bits))
        v_p = q.variance_pauli_all(bits, [1] * len(bits))
        assert v_u == pytest.approx(v_p, abs=1e-9)
    # single-bit: sum and product observables coincide
    assert q.expectation_pauli_all([2], [1]) == pytest.approx(
        q.pauli_expectation([2], [1]), abs=1e-9)


def test_pauli_product_on_bell():
    q = qa.create_simulator(2, engine="cpu", precision="fp64", seed=1)
    q.h(0)
    q.cnot(0, 1)
    # product <Z x Z> = 1 on Bell, by both the in-place parity path and the
    # reference's basis-rotated factorized-product path
    assert q.pauli_expectation([0, 1], [2, 2]) == pytest.approx(1.0, abs=1e-9)
    assert q.expectation_pauli_all([0, 1], [2, 2]) == pytest.approx(1.0, abs=1e-9)
    assert q.expectation_pauli_all([0, 1], [1, 1]) == pytest.approx(1.0, abs=1e-9)
    assert q.expectation_pauli_all([0, 1], [3, 3]) == pytest.approx(-1.0, abs=1e-9)
    # the product observable squares to identity: Var = 1 - E^2 = 0 here
    assert q.variance_pauli_all([0, 1], [2, 2]) == pytest.approx(0.0, abs=1e-9)


def test_unitary_all_custom_eigenvalues():
    q, sv = make_state()
    ident = [1, 0, 0, 1]
    p1 = abs(sv[np.arange(16) & 1 == 1]).sum()  # placeholder, recompute below
    p1 = float((np.abs(sv) ** 2)[(np.arange(16) & 1) == 1].sum())
    e = q.expectation_unitary_all([0], ident, [2.0, 5.0])
    assert e == pytest.approx(2.0 * (1 - p1) + 5.0 * p1, abs=1e-9)


def test_rdm_aliases_cpp_layer():
    # Rdm forms equal exact forms on exact states (checked through capi-free
    # python: the binding exposes only exact forms; exercise layered sim)
    q = qa.create_simulator(3, layers=["qunit", "cpu"], seed=1)
    q.h(0)
    q.cnot(0, 1)
    bits = [0, 1]
    jp = np.asarray(q.prob_bits_all(bits))
    np.testing.assert_allclose(jp, [0.5, 0, 0, 0.5], atol=1e-6)
