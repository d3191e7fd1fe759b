"""Extended named-gate / rotation / boolean-logic / shift / adder API
(parity: reference rotational.cpp, gates.cpp, logic.cpp, arithmetic.cpp)."""

import numpy as np
import pytest

import qrack_amd as qa
from ref_sim import assert_states_close


def cpu(n, seed=1, precision="fp64"):
    return qa.create_simulator(n, engine="cpu", seed=seed, precision=precision)


def test_ai_iai_roundtrip():
    q = cpu(1)
    q.ai(0, 0.7, 1.1)
    assert abs(q.prob(0) - np.sin(1.1 / 2) ** 2) < 1e-9
    q.iai(0, 0.7, 1.1)
    assert abs(q.prob(0)) < 1e-9


def test_controlled_ai():
    q = cpu(2)
    q.x(1)
    q.cai(1, 0, 0.3, 0.9)
    assert abs(q.prob(0) - np.sin(0.9 / 2) ** 2) < 1e-9
    q.ciai(1, 0, 0.3, 0.9)
    assert abs(q.prob(0)) < 1e-9
    q2 = cpu(2)
    q2.anti_cai(1, 0, 0.3, 0.9)  # control |0>: fires
    assert abs(q2.prob(0) - np.sin(0.9 / 2) ** 2) < 1e-9
    q2.anti_ciai(1, 0, 0.3, 0.9)
    assert abs(q2.prob(0)) < 1e-9


def test_named_composites_square_to_parents():
    # SqrtH^2 == H, SqrtW * ISqrtW == I, SH == S*H, HIS == H*IS
    q1, q2 = cpu(1, 3), cpu(1, 3)
    q1.sqrt_h(0)
    q1.sqrt_h(0)
    q2.h(0)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-9)
    q3 = cpu(1)
    q3.sqrt_w(0)
    q3.isqrt_w(0)
    assert abs(q3.get_amplitude(0) - 1) < 1e-9
    q4, q5 = cpu(1, 5), cpu(1, 5)
    q4.sh(0)
    q5.h(0)
    q5.s(0)
    assert_states_close(q4.get_state_vector(), q5.get_state_vector(), 1e-9)
    q6, q7 = cpu(1, 6), cpu(1, 6)
    q6.h(0)
    q6.sh(0)
    q6.his(0)
    q7.h(0)
    assert_states_close(q6.get_state_vector(), q7.get_state_vector(), 1e-9)


def test_u2_iu2_roundtrip():
    q = cpu(1)
    q.u2(0, 0.4, 1.3)
    q.iu2(0, 0.4, 1.3)
    assert abs(q.get_amplitude(0) - 1) < 1e-9


def test_exp_family():
    # ExpZ on |0> = phase e^{i r}; ExpX on |0> = e^{i r} |1>
    q = cpu(1)
    q.exp_z(0.5, 0)
    a = complex(q.get_amplitude(0))
    assert abs(a - np.exp(0.5j)) < 1e-9
    q2 = cpu(1)
    q2.exp_x(0.3, 0)
    a1 = complex(q2.get_amplitude(1))
    assert abs(a1 - np.exp(0.3j)) < 1e-9


def test_dyads_equal_angles():
    q1, q2 = cpu(1, 7), cpu(1, 7)
    q1.ry_dyad(3, 4, 0)
    q2.ry(-2 * np.pi * 3 / 16, 0)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-9)


def test_crx_cry_crt():
    q = cpu(2)
    q.x(1)
    q.cry(1.1, 1, 0)
    assert abs(q.prob(0) - np.sin(1.1 / 2) ** 2) < 1e-9
    q2 = cpu(2)
    q2.crx(0.8, 1, 0)  # control |0>: no-op
    assert abs(q2.prob(0)) < 1e-12


def test_uniformly_controlled_ry_rz():
    angles = [0.3, 0.9]
    q = cpu(2)
    q.x(1)
    q.uniformly_controlled_ry([1], 0, angles)
    assert abs(q.prob(0) - np.sin(0.9 / 2) ** 2) < 1e-9
    q2 = cpu(2)
    q2.uniformly_controlled_ry([1], 0, angles)
    assert abs(q2.prob(0) - np.sin(0.3 / 2) ** 2) < 1e-9


def test_uc_phase_invert():
    q = cpu(2)
    q.h(0)
    q.uc_invert([1], 1, 1, 0, 0)  # fires when control is |0>
    # X after H: |+> unchanged
    assert abs(q.prob(0) - 0.5) < 1e-9
    q.uc_phase([1], 1, -1, 0, 0)  # Z when control |0>
    q.h(0)
    assert abs(q.prob(0) - 1.0) < 1e-9  # HZH|+> = |1>


def test_boolean_logic_truth_tables():
    for a in (0, 1):
        for b in (0, 1):
            for op, expect in (("and_", a & b), ("or_", a | b), ("xor_", a ^ b),
                               ("nand", 1 - (a & b)), ("nor", 1 - (a | b)),
                               ("xnor", 1 - (a ^ b))):
                q = cpu(3)
                if a:
                    q.x(0)
                if b:
                    q.x(1)
                getattr(q, op)(0, 1, 2)
                assert round(q.prob(2)) == expect, (op, a, b)


def test_classical_logic():
    for a in (0, 1):
        for c in (False, True):
            for op, expect in (("cland", a & c), ("clor", a | c), ("clxor", a ^ c),
                               ("clnand", 1 - (a & c)), ("clnor", 1 - (a | c)),
                               ("clxnor", 1 - (a ^ c))):
                q = cpu(2)
                if a:
                    q.x(0)
                getattr(q, op)(0, bool(c), 1)
                assert round(q.prob(1)) == int(expect), (op, a, c)


def test_shifts():
    q = cpu(6)
    q.set_reg(0, 5, 0b00101)
    q.lsl(1, 0, 5)
    assert q.m_reg(0, 5) == 0b01010
    q.lsr(1, 0, 5)
    assert q.m_reg(0, 5) == 0b00101
    # ASL/ASR treat the top two bits as sign and carry (reference
    # qinterface.cpp:335-368 wiring: park sign next door, rotate, zero-fill)
    q2 = cpu(6)
    q2.set_reg(0, 5, 0b10011)
    q2.asl(1, 0, 5)
    assert q2.m_reg(0, 5) == 0b01110
    q2.asr(1, 0, 5)
    assert q2.m_reg(0, 5) == 0b00011


def test_set_bit_set_reg_reverse():
    q = cpu(5)
    q.set_bit(3, True)
    assert q.m(3) is True
    q.set_reg(0, 4, 0b0110)
    assert q.m_reg(0, 4) == 6
    q.reverse(0, 4)
    assert q.m_reg(0, 4) == 0b0110  # 0110 reversed = 0110


def test_reverse_asymmetric():
    q = cpu(4)
    q.set_reg(0, 4, 0b0001)
    q.reverse(0, 4)
    assert q.m_reg(0, 4) == 0b1000


def test_adc_iadc():
    # [in1](3) + [in2](3) -> output(3)+carry, inputs preserved
    q = cpu(10)
    q.set_reg(0, 3, 5)
    q.set_reg(3, 3, 6)
    q.adc(0, 3, 6, 3, 9)
    out = q.m_reg(6, 3) | (int(q.m(9)) << 3)
    assert out == 11
    assert q.m_reg(0, 3) == 5 and q.m_reg(3, 3) == 6
    q.iadc(0, 3, 6, 3, 9)
    assert q.m_reg(6, 3) == 0 and q.m(9) is False


def test_cadc_controlled():
    for ctl in (0, 1):
        q = cpu(11)
        if ctl:
            q.x(10)
        q.set_reg(0, 3, 3)
        q.set_reg(3, 3, 2)
        q.cadc([10], 0, 3, 6, 3, 9)
        assert q.m_reg(6, 3) == (5 if ctl else 0)
        q.ciadc([10], 0, 3, 6, 3, 9)
        assert q.m_reg(6, 3) == 0


def test_phase_root_n_mask():
    q1, q2 = cpu(3, 9), cpu(3, 9)
    for i in range(3):
        q1.h(i)
        q2.h(i)
    q1.phase_root_n_mask(2, 0b101)
    q2.phase_root_n(2, 0)
    q2.phase_root_n(2, 2)
    assert_states_close(q1.get_state_vector(), q2.get_state_vector(), 1e-9)


def test_anti_controlled_named():
    q = cpu(2)
    q.anti_cs(1, 0)  # control |0>: S on target |0> = no phase visible
    q.h(0)
    q.anti_ct(1, 0)
    q.anti_cit(1, 0)
    q.h(0)
    assert abs(q.prob(0)) < 1e-9
    q2 = cpu(3)
    q2.x(0)
    q2.ccy(0, 1, 2)  # c2 = |0>: no-op
    assert abs(q2.prob(2)) < 1e-12
    q2.x(1)
    q2.ccy(0, 1, 2)
    assert abs(q2.prob(2) - 1.0) < 1e-9


def test_misc_knobs_and_queries():
    q = cpu(3)
    q.set_concurrency(4)  # hint; no-op
    q.x(1)
    assert q.highest_prob_all() == 2
    q.h(0)
    # first nonzero phase of H|0> (x) |010> is 0
    assert abs(q.first_nonzero_phase()) < 1e-9
    s = q.sample_clone([1, 2, 4])
    assert s in (0b010, 0b011)
    # sampling from a clone must not disturb the state
    assert abs(q.prob(0) - 0.5) < 1e-9


def test_noise_parameter_knob():
    q = qa.create_simulator(2, layers=["noisy", "cpu"], seed=1)
    q.set_noise_parameter(0.0)
    assert q.get_noise_parameter() == 0.0
    q.h(0)
    q.cnot(0, 1)
    assert abs(q.prob(1) - 0.5) < 1e-6  # zero noise: exact Bell


def test_ace_max_qubits_knob():
    q = qa.create_simulator(8, layers=["qunit", "cpu"], seed=2)
    q.set_ace_max_qubits(2)
    assert q.get_ace_max_qubits() == 2
    q.h(0)
    q.cnot(0, 1)  # exactly at cap: allowed
    assert abs(q.prob(1) - 0.5) < 1e-6


def test_try_decompose():
    # separable split succeeds; entangled split fails and leaves state intact
    q = cpu(3)
    q.x(2)
    dest = qa.create_simulator(1, engine="cpu", precision="fp64", seed=9)
    assert q.try_decompose(2, dest)
    assert q.num_qubits == 2
    assert abs(dest.prob(0) - 1.0) < 1e-9
    q2 = cpu(2)
    q2.h(0)
    q2.cnot(0, 1)
    dest2 = qa.create_simulator(1, engine="cpu", precision="fp64", seed=9)
    assert not q2.try_decompose(1, dest2)
    assert q2.num_qubits == 2
    assert abs(q2.prob(1) - 0.5) < 1e-9


def test_are_factorized():
    q = qa.create_simulator(4, layers=["qunit", "cpu"], seed=3)
    q.h(0)
    q.cnot(0, 1)
    assert not q.are_factorized([0], [1])
    assert q.are_factorized([0, 1], [2, 3])
    assert qa.create_simulator(2, engine="cpu").are_factorized([0], [1]) is False


def test_sparse_knobs():
    q = qa.create_simulator(10, engine="sparse", seed=4)
    q.set_sparse_probability_floor(1e-10)
    q.set_sparse_ace_max_mb(64)
    for i in range(10):
        q.h(i)
    assert abs(q.prob(0) - 0.5) < 1e-4
